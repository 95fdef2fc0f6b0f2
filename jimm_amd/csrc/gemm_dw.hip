// K15 — split-K TN GEMM for weight gradients: dW[N,K] = dz^T @ x.
//
// rocBLAS/Tensile picks non-split kernels for these shapes (output tiles
// 27-144 workgroups on a 256-CU chip) and measures 230-578 TF/s
// (benchmarks/bwd_gemm_bench.py, profiles/r01_NOTES.md). This kernel
// splits the huge contraction dim (M = B*L ~ 50k) across S slices so the
// grid is (N/128)*(K/128)*S workgroups, with fp32 atomicAdd reduction.
//
//   * 128x128 output tile, 4 waves (2x2), 64x64 per wave
//     (4x4 fragments of v_mfma_f32_16x16x32_bf16);
//   * both operands scatter-transposed into LDS per 64-row m-chunk
//     (dz^T[n][m], x^T[k][m]) so A and B fragments are contiguous
//     ds_read_b128 (same idiom as the attention kernels' V^T staging);
//   * m-chunks are slice-INTERLEAVED (chunk c of slice s is m-block
//     c*S+s) so concurrently-running slices sweep the same ~1/S band of
//     dz and x — the band stays LLC-resident instead of streaming 300 MB;
//   * fp32 atomicAdd epilogue into a zeroed (N,K) workspace; host casts
//     to bf16.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BN = 128;   // dW rows (columns of dz)
constexpr int BKC = 128;  // dW cols (columns of x)
constexpr int BM = 64;    // contraction chunk (rows of dz/x)
constexpr int PITCH = BM + 8;

__global__ __launch_bounds__(256) void gemm_tn_splitk_kernel(
    const bf16* __restrict__ dz, const bf16* __restrict__ x, float* __restrict__ dw,
    int M, int N, int K, int S) {
  // LDS: dz^T [128][72], x^T [128][72] shorts = 36.9 KiB (single buffer)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* dzt = reinterpret_cast<short*>(smem);
  short* xt = dzt + BN * PITCH;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  const int kt = K / BKC;
  const int tile = blockIdx.x / S;
  const int s = blockIdx.x % S;
  const int n0 = (tile / kt) * BN;
  const int k0 = (tile % kt) * BKC;
  const int wn = (wave >> 1) * 64;   // within-tile n offset
  const int wk = (wave & 1) * 64;    // within-tile k offset

  f32x4_t acc[4][4] = {};

  const int nchunks = M / BM;
  for (int c = s; c < nchunks; c += S) {
    const int m0 = c * BM;
    // ---- scatter-transpose staging: thread (m=tid/4, 32-col chunk) -------
    {
      const int m = tid / 4;
      const int c0 = (tid % 4) * 32;
#pragma unroll
      for (int h = 0; h < 4; ++h) {
        const bf16x8_t d8 =
            *reinterpret_cast<const bf16x8_t*>(dz + (int64_t)(m0 + m) * N + n0 + c0 + h * 8);
        const bf16x8_t x8 =
            *reinterpret_cast<const bf16x8_t*>(x + (int64_t)(m0 + m) * K + k0 + c0 + h * 8);
#pragma unroll
        for (int i = 0; i < 8; ++i) dzt[(c0 + h * 8 + i) * PITCH + m] = d8[i];
#pragma unroll
        for (int i = 0; i < 8; ++i) xt[(c0 + h * 8 + i) * PITCH + m] = x8[i];
      }
    }
    __syncthreads();

    // ---- 32 MFMA per wave over the 64-deep chunk -------------------------
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t a[4], b[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        a[mi] = *reinterpret_cast<const bf16x8_t*>(dzt + (wn + 16 * mi + lo) * PITCH + 32 * ks + hi * 8);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        b[ni] = *reinterpret_cast<const bf16x8_t*>(xt + (wk + 16 * ni + lo) * PITCH + 32 * ks + hi * 8);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = MFMA16(a[mi], b[ni], acc[mi][ni]);
    }
    __syncthreads();
  }

  // ---- fp32 atomic reduction into the workspace ---------------------------
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int n = n0 + wn + 16 * mi + hi * 4 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int k = k0 + wk + 16 * ni + lo;
        atomicAdd(dw + (int64_t)n * K + k, acc[mi][ni][r]);
      }
    }
  }
}

}  // namespace

bool gemm_dw_supported(int64_t M, int64_t N, int64_t K) {
  return (N % BN == 0) && (K % BKC == 0) && (M % BM == 0) && M >= BM;
}

torch::Tensor gemm_tn_splitk(torch::Tensor dz, torch::Tensor x) {
  // dz (M,N), x (M,K) bf16 contiguous -> dW (N,K) bf16
  TORCH_CHECK(dz.is_cuda() && dz.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dz.scalar_type() == torch::kBFloat16 && x.scalar_type() == torch::kBFloat16);
  const int M = dz.size(0), N = dz.size(1), K = x.size(1);
  TORCH_CHECK(x.size(0) == M && gemm_dw_supported(M, N, K));
  auto ws = torch::zeros({N, K}, dz.options().dtype(torch::kFloat32));
  const int tiles = (N / BN) * (K / BKC);
  int S = (512 + tiles - 1) / tiles;           // target ~512-1024 workgroups
  S = std::max(1, std::min({S, 32, M / BM}));
  const size_t shmem = 2 * BN * PITCH * sizeof(short);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(gemm_tn_splitk_kernel, dim3(tiles * S), dim3(256), shmem, stream,
                     reinterpret_cast<const bf16*>(dz.data_ptr()),
                     reinterpret_cast<const bf16*>(x.data_ptr()), ws.data_ptr<float>(),
                     M, N, K, S);
  return ws.to(torch::kBFloat16);
}

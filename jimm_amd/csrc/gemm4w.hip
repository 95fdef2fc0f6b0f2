// Experimental 4-wave NT GEMM (one wave per SIMD, Tensile-like structure).
//
// Hypothesis (from the r02 PMC profiles): the 8-wave lockstep phase
// schedule of gemm8p.hip parks 26-29% of wave time at barriers/vmcnt;
// Tensile's kernels run ~86% MFMA-busy with big per-wave tiles and almost
// no synchronization. This kernel: 256 threads = 4 waves (2x2), each wave
// owns a 128x128 output (8x8 MFMA fragments, 256 accumulator registers —
// the unified VGPR/AGPR file allows ~512 at 1 wave/SIMD), BK=32 K-tiles
// cycling over FOUR LDS slots per operand so the glds pipeline stays two
// tiles deep with a single raw barrier + counted vmcnt(8) per K-tile:
//   iter t: [16 fragment ds_reads (tile t) | stage tile t+2 (8 glds/wave) |
//            64 MFMA | vmcnt(8) | s_barrier]
// Certification: reads(t) follow barrier(t-1) which follows every wave's
// vmcnt(8) = "all but tile t+1's stages landed" => tile t resident.  Slot
// reuse (mod 4) overwrites tile t-2's image, whose readers finished >= 2
// barriers earlier.  LDS image swizzle identical to gemm8p.hip.
//
// Epilogue: bias only (A/B experiment; promoted to the full epilogue set
// if it beats gemm8p on the model shapes).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BM = 256, BN = 256, BK = 32;
constexpr int NTHREADS = 256;
constexpr int IMG = BM * BK;  // shorts per operand slot (16 KiB)

__device__ __forceinline__ void glds16w(const bf16* g, char* lds_dst) {
  typedef const __attribute__((address_space(1))) unsigned int* gp_t;
  typedef __attribute__((address_space(3))) unsigned int* lp_t;
  __builtin_amdgcn_global_load_lds((gp_t)(const void*)g, (lp_t)(void*)lds_dst, 16, 0, 0);
}

// image byte(row, c16) = row*64 + (c16 ^ (row&3))*16  — BK=32: 4 chunks/row
__device__ __forceinline__ int frag_off4(int row, int chunk) {
  return (row << 6) + ((chunk ^ (row & 3)) << 4);
}

template <bool HAS_BIAS, bool MGUARD>
__global__ __launch_bounds__(NTHREADS, 1) void gemm_nt_4w_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W, const float* __restrict__ bias,
    bf16* __restrict__ Y, int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto As = [&](int slot) { return smem + slot * (IMG * 2); };
  auto Bs = [&](int slot) { return smem + 4 * IMG * 2 + slot * (IMG * 2); };

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  const int mt = MGUARD ? (M + BM - 1) / BM : M / BM;
  const int nt = N / BN;
  const int nwg = mt * nt;
  int wg = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg / nt) * BM;
  const int n0 = (wg % nt) * BN;
  const int wm = (wave >> 1) * 128;  // 2x2 wave grid, 128x128 per wave
  const int wn = (wave & 1) * 128;

  f32x4_t acc[8][8] = {};  // [mi][ni]

  const int nk = K / BK;
  // staging: 256 threads x 16 B x 4 rounds per 16 KiB image; per-lane
  // source base with the read swizzle inverted (rule 21).  A stage() is
  // 8 glds per wave (4 rounds x 2 operands).
  const bf16* asrc[4];
  const bf16* bsrc[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int off = r * 4096 + tid * 16;
    const int row = off >> 6;
    const int chunk = ((off >> 4) & 3) ^ (row & 3);
    int ga = m0 + row;
    if (MGUARD) ga = ga < M ? ga : M - 1;
    asrc[r] = X + (int64_t)ga * K + chunk * 8;
    bsrc[r] = W + (int64_t)(n0 + row) * K + chunk * 8;
  }
  const int ldst = tid * 16;
  auto stage = [&](int kt) {
    const int slot = kt & 3;
    char* ai = As(slot);
    char* bi = Bs(slot);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      glds16w(asrc[r] + (int64_t)kt * BK, ai + r * 4096 + ldst);
      glds16w(bsrc[r] + (int64_t)kt * BK, bi + r * 4096 + ldst);
    }
  };

  int offA[8], offB[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    offA[i] = frag_off4(wm + 16 * i + lo, hi);
    offB[i] = frag_off4(wn + 16 * i + lo, hi);
  }

  // prologue: stage tiles 0 and 1; certify tile 0 (a stage = 8 glds/wave)
  stage(0);
  if (nk > 1) {
    stage(1);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  asm volatile("s_barrier" ::: "memory");

  for (int t = 0; t < nk; ++t) {
    const int slot = t & 3;
    const char* ai = As(slot);
    const char* bi = Bs(slot);
    bf16x8_t xa[8], wb[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) xa[i] = *reinterpret_cast<const bf16x8_t*>(ai + offA[i]);
#pragma unroll
    for (int i = 0; i < 8; ++i) wb[i] = *reinterpret_cast<const bf16x8_t*>(bi + offB[i]);
    if (t + 2 < nk) stage(t + 2);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mi = 0; mi < 8; ++mi)
#pragma unroll
      for (int ni = 0; ni < 8; ++ni) acc[mi][ni] = MFMA16(xa[mi], wb[ni], acc[mi][ni]);
    __builtin_amdgcn_s_setprio(0);
    if (t + 1 < nk) {
      // allow only this iter's stage (tile t+2, 8 loads) to stay in flight;
      // when the stream has ended the drain certifies the last tiles
      if (t + 2 < nk) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      asm volatile("s_barrier" ::: "memory");
    }
  }

  // epilogue
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wm + 16 * mi + hi * 4 + r;
      if (MGUARD && m >= M) continue;
#pragma unroll
      for (int ni = 0; ni < 8; ++ni) {
        const int n = n0 + wn + 16 * ni + lo;
        float v = acc[mi][ni][r];
        if (HAS_BIAS) v += bias[n];
        Y[(int64_t)m * N + n] = f2bf(v);
      }
    }
  }
}

}  // namespace

torch::Tensor gemm_nt_4w(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(N % BN == 0 && K % BK == 0 && K >= 2 * BK);
  auto y = torch::empty({M, N}, x.options());
  c10::optional<torch::Tensor> bf;
  if (bias) bf = bias->contiguous().to(torch::kFloat32);
  const float* biasp = bf ? bf->data_ptr<float>() : nullptr;
  auto stream = at::hip::getCurrentHIPStream();
  const size_t shmem = 8 * IMG * 2;  // 128 KiB
  const bool mguard = (M % BM) != 0;
  const int mt = (M + BM - 1) / BM;
#define L4W(HB, MG)                                                                        \
  do {                                                                                    \
    auto kfn = gemm_nt_4w_kernel<HB, MG>;                                                 \
    static bool attr_##HB##MG = [&] {                                                     \
      hipFuncSetAttribute(reinterpret_cast<const void*>(kfn),                             \
                          hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);        \
      return true;                                                                        \
    }();                                                                                  \
    (void)attr_##HB##MG;                                                                  \
    hipLaunchKernelGGL(kfn, dim3(mt * (N / BN)), dim3(NTHREADS), shmem, stream,           \
                       reinterpret_cast<const bf16*>(x.data_ptr()),                       \
                       reinterpret_cast<const bf16*>(w.data_ptr()), biasp,                \
                       reinterpret_cast<bf16*>(y.data_ptr()), M, N, K);                   \
  } while (0)
  const bool hb = bias.has_value();
  if (hb && mguard) L4W(true, true);
  else if (hb) L4W(true, false);
  else if (mguard) L4W(false, true);
  else L4W(false, false);
#undef L4W
  return y;
}

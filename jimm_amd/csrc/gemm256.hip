// K4/K6/K7/K8 — 256x256-tile MFMA bf16 GEMM with fused epilogue (gfx950).
//
// The 128x128-tile kernel (gemm.hip) measured 0.5-0.8 PF/s on the ViT-B
// training shapes (profiles/r01_NOTES.md): at 64 FLOP/byte of HBM traffic a
// 128x128 macro-tile is bandwidth-bound well below the MFMA rate. This
// kernel implements the CDNA4 guide's verified 256-square structure
// (cdna_hip_programming.md §5 "glds, 2 LDS buffers, BK=64, vmcnt(0) +
// plain __syncthreads()" — measured top-tier ~1.2 PF/s at 4096^3):
//   * 256x256 macro-tile, BK=64, 8 waves (2M x 4N), 128x64 per wave
//     (8x4 fragments of v_mfma_f32_16x16x32_bf16);
//   * both operand tiles staged by __builtin_amdgcn_global_load_lds
//     width 16, double-buffered: 4 x 32 KiB = 128 KiB LDS;
//   * LDS image XOR-swizzled exactly as gemm.hip (128-B rows, 16-B chunk
//     ^= row&7, inverse swizzle on the per-lane glds SOURCE address);
//   * XCD-aware bijective tile remap (same formula as gemm.hip).
// Epilogue fuses bias + gelu/quickgelu (+residual) like gemm.hip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTHREADS = 512;

// LDS linear image per operand tile: [256 rows][64 shorts] = 128 B/row,
// 32 KiB; chunk c (16 B) of row r lives at byte r*128 + (c ^ (r&7))*16.

__device__ __forceinline__ void stage256_glds(const bf16* __restrict__ gsrc, int64_t ldg,
                                              short* lds_base, int tid) {
  // 512 threads stage 256x64 shorts (32 KiB): 4 rounds of 8 KiB.
#pragma unroll
  for (int round = 0; round < 4; ++round) {
    const int off = round * 8192 + tid * 16;  // byte offset in image
    const int row = off >> 7;                 // /128
    const int chunk = (off >> 4) & 7;
    const int src_chunk = chunk ^ (row & 7);
    const bf16* g = gsrc + (int64_t)row * ldg + src_chunk * 8;
    typedef const __attribute__((address_space(1))) unsigned int* gp_t;
    typedef __attribute__((address_space(3))) unsigned int* lp_t;
    __builtin_amdgcn_global_load_lds((gp_t)(const void*)g,
                                     (lp_t)(void*)(reinterpret_cast<char*>(lds_base) + off),
                                     16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8_t lds_frag256(const short* base, int row, int chunk) {
  const int byte = (row << 7) + ((chunk ^ (row & 7)) << 4);
  return *reinterpret_cast<const bf16x8_t*>(reinterpret_cast<const char*>(base) + byte);
}

template <int ACT, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE>
__global__ __launch_bounds__(NTHREADS, 2) void gemm_nt_256_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W, const float* __restrict__ bias,
    const bf16* __restrict__ res, bf16* __restrict__ Y, bf16* __restrict__ Z,
    int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* smem_s = reinterpret_cast<short*>(smem);
  // buffers: X tiles at [0,16384) and [16384,32768); W tiles at +32768
  auto xs = [&](int buf) { return smem_s + buf * 16384; };
  auto ws = [&](int buf) { return smem_s + 32768 + buf * 16384; };

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  // XCD-aware bijective remap of the linear tile id
  const int mt = M / BM, nt = N / BN;
  const int nwg = mt * nt;
  int wg = blockIdx.x;
  {
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg / nt) * BM;
  const int n0 = (wg % nt) * BN;
  const int wm = (wave >> 2) * 128;  // 2 M-waves
  const int wn = (wave & 3) * 64;    // 4 N-waves

  f32x4_t acc[8][4] = {};

  stage256_glds(X + (int64_t)m0 * K, K, xs(0), tid);
  stage256_glds(W + (int64_t)n0 * K, K, ws(0), tid);
  __syncthreads();

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nk) {
      stage256_glds(X + (int64_t)m0 * K + (kt + 1) * BK, K, xs(buf ^ 1), tid);
      stage256_glds(W + (int64_t)n0 * K + (kt + 1) * BK, K, ws(buf ^ 1), tid);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t xa[8], wb[4];
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) xa[mi] = lds_frag256(xs(buf), wm + 16 * mi + lo, 4 * ks + hi);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) wb[ni] = lds_frag256(ws(buf), wn + 16 * ni + lo, 4 * ks + hi);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = MFMA16(xa[mi], wb[ni], acc[mi][ni]);
    }
    __syncthreads();  // drains in-flight global_load_lds (vmcnt 0) + tile reuse
  }

  // epilogue: C rows = m (hi*4+r per 16-fragment), col = n (lo)
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wm + 16 * mi + hi * 4 + r;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int n = n0 + wn + 16 * ni + lo;
        float vpre = acc[mi][ni][r];
        if (HAS_BIAS) vpre += bias[n];
        if (SAVE_PRE) Z[(int64_t)m * N + n] = f2bf(vpre);
        float vy = act_fwd(vpre, ACT);
        if (HAS_RES) vy += bf2f(res[(int64_t)m * N + n]);
        Y[(int64_t)m * N + n] = f2bf(vy);
      }
    }
  }
}

}  // namespace

bool gemm256_supported(int64_t M, int64_t N, int64_t K) {
  return (M % BM == 0) && (N % BN == 0) && (K % BK == 0);
}

void gemm_nt_256(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias_f32,
                 std::string act, c10::optional<torch::Tensor> residual, torch::Tensor y,
                 c10::optional<torch::Tensor> z) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(gemm256_supported(M, N, K));
  int act_code = ACT_NONE;
  if (act == "gelu") act_code = ACT_GELU;
  else if (act == "gelu_tanh") act_code = ACT_GELU_TANH;
  else if (act == "quickgelu") act_code = ACT_QUICKGELU;
  else TORCH_CHECK(act.empty(), "unknown act ", act);
  auto stream = at::hip::getCurrentHIPStream();
  const bf16* resp = residual ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
  const float* biasp = bias_f32 ? bias_f32->data_ptr<float>() : nullptr;
  bf16* zp = z ? reinterpret_cast<bf16*>(z->data_ptr()) : nullptr;
  const size_t shmem = 4 * 16384 * sizeof(short);  // 128 KiB

#define LAUNCH256(ACTC, HB, HR, SP)                                                        \
  do {                                                                                     \
    auto kfn = gemm_nt_256_kernel<ACTC, HB, HR, SP>;                                       \
    static bool attr_set_##ACTC##HB##HR##SP = [&] {                                        \
      hipFuncSetAttribute(reinterpret_cast<const void*>(kfn),                              \
                          hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);         \
      return true;                                                                         \
    }();                                                                                   \
    (void)attr_set_##ACTC##HB##HR##SP;                                                     \
    hipLaunchKernelGGL(kfn, dim3((M / BM) * (N / BN)), dim3(NTHREADS), shmem, stream,      \
                       reinterpret_cast<const bf16*>(x.data_ptr()),                        \
                       reinterpret_cast<const bf16*>(w.data_ptr()), biasp, resp,           \
                       reinterpret_cast<bf16*>(y.data_ptr()), zp, M, N, K);                \
  } while (0)
#define DISPATCH_ACT256(HB, HR, SP)                                                        \
  switch (act_code) {                                                                      \
    case ACT_NONE: LAUNCH256(ACT_NONE, HB, HR, SP); break;                                 \
    case ACT_GELU: LAUNCH256(ACT_GELU, HB, HR, SP); break;                                 \
    case ACT_GELU_TANH: LAUNCH256(ACT_GELU_TANH, HB, HR, SP); break;                       \
    case ACT_QUICKGELU: LAUNCH256(ACT_QUICKGELU, HB, HR, SP); break;                       \
  }
  const bool hb = bias_f32.has_value(), hr = residual.has_value(), sp = z.has_value();
  if (hb && hr && sp) DISPATCH_ACT256(true, true, true)
  else if (hb && hr) DISPATCH_ACT256(true, true, false)
  else if (hb && sp) DISPATCH_ACT256(true, false, true)
  else if (hb) DISPATCH_ACT256(true, false, false)
  else if (hr && sp) DISPATCH_ACT256(false, true, true)
  else if (hr) DISPATCH_ACT256(false, true, false)
  else if (sp) DISPATCH_ACT256(false, false, true)
  else DISPATCH_ACT256(false, false, false)
#undef DISPATCH_ACT256
#undef LAUNCH256
}

// K4/K6/K7/K8 — MFMA bf16 GEMM with fused bias/activation/residual epilogue.
//
// linear_fwd computes y = act(x @ w^T + bias) [+ residual] for x (M,K)
// row-major and w (N,K) row-major (torch Linear convention) — an "NT" GEMM
// mapped onto v_mfma_f32_16x16x32_bf16 tiles.
//
// v1 structure (CDNA-guide §5 ladder step ~2): 128x128 macro-tile, 4 waves,
// each wave a 64x64 sub-tile of 4x4 16x16 fragments; K staged in LDS
// double-buffered; epilogue applies bias/act/residual in-register before the
// bf16 store. The 256²/8-phase schedule (guide §5 template) is the planned
// upgrade once the baseline is profiled.
//
// gemm_supported() gates dispatch: the python layer falls back to
// rocBLAS (torch.matmul) + the fused bias_act kernel when unsupported.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int PITCH = BK + 8;  // shorts; pad keeps b128 row reads conflict-light

// act codes shared with elementwise.hip via common.h
template <int ACT, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE>
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W, const float* __restrict__ bias,
    const bf16* __restrict__ res, bf16* __restrict__ Y, bf16* __restrict__ Z,
    int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* xs = reinterpret_cast<short*>(smem);              // [2][BM][PITCH]
  short* ws = xs + 2 * BM * PITCH;                         // [2][BN][PITCH]

  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  const int m0 = blockIdx.x * BM;  // block row
  const int n0 = blockIdx.y * BN;  // block col
  // wave sub-tile: 2x2 wave grid, each wave 64x64
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;

  f32x4_t acc[4][4] = {};  // [mi][ni] 16x16 fragments

  auto stage = [&](int buf, int k0) {
    // 256 threads stage BM x BK of X and BN x BK of W (bf16, row-major K-major)
    // each thread: 128*64/256 = 32 shorts of each = 2 x bf16x8 x 2
    short* xd = xs + buf * BM * PITCH;
    short* wd = ws + buf * BN * PITCH;
    const int row = tid / 2;           // 0..127
    const int c0 = (tid & 1) * 32;     // two 32-short halves
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cc = c0 + h * 16;
      // X rows are guarded (M may be ragged); W rows assumed N%? guarded too
      if (m0 + row < M) {
        *reinterpret_cast<bf16x8_t*>(xd + row * PITCH + cc) =
            *reinterpret_cast<const bf16x8_t*>(X + (int64_t)(m0 + row) * K + k0 + cc);
        *reinterpret_cast<bf16x8_t*>(xd + row * PITCH + cc + 8) =
            *reinterpret_cast<const bf16x8_t*>(X + (int64_t)(m0 + row) * K + k0 + cc + 8);
      } else {
        for (int i = 0; i < 16; ++i) xd[row * PITCH + cc + i] = 0;
      }
      if (n0 + row < N) {
        *reinterpret_cast<bf16x8_t*>(wd + row * PITCH + cc) =
            *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + row) * K + k0 + cc);
        *reinterpret_cast<bf16x8_t*>(wd + row * PITCH + cc + 8) =
            *reinterpret_cast<const bf16x8_t*>(W + (int64_t)(n0 + row) * K + k0 + cc + 8);
      } else {
        for (int i = 0; i < 16; ++i) wd[row * PITCH + cc + i] = 0;
      }
    }
  };

  stage(0, 0);
  __syncthreads();

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nk) stage(buf ^ 1, (kt + 1) * BK);
    short* xd = xs + buf * BM * PITCH;
    short* wd = ws + buf * BN * PITCH;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {  // two 32-deep steps per BK
      bf16x8_t xa[4], wb[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        xa[mi] = *reinterpret_cast<const bf16x8_t*>(xd + (wm + 16 * mi + lo) * PITCH + 32 * ks + hi * 8);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        wb[ni] = *reinterpret_cast<const bf16x8_t*>(wd + (wn + 16 * ni + lo) * PITCH + 32 * ks + hi * 8);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = MFMA16(xa[mi], wb[ni], acc[mi][ni]);
    }
    __syncthreads();
  }

  // epilogue: C fragment rows = x rows (m), cols = w rows (n)
  // A=x fragment rows are A[l&15] -> m index; B=w fragment cols l&15 -> n.
  // C[i][j]: lane holds rows (hi*4+r) of m-tile, col lo of n-tile.
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wm + 16 * mi + hi * 4 + r;
      if (m >= M) continue;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int n = n0 + wn + 16 * ni + lo;
        if (n >= N) continue;
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

bool gemm_supported(int64_t M, int64_t N, int64_t K, std::string dtype) {
  if (dtype != "torch.bfloat16") return false;
  if (K % BK != 0) return false;
  const char* env = getenv("JIMM_AMD_GEMM");
  if (env && std::string(env) == "blas") return false;
  return true;
}

std::vector<torch::Tensor> linear_fwd(torch::Tensor x, torch::Tensor w,
                                      c10::optional<torch::Tensor> bias, std::string act,
                                      c10::optional<torch::Tensor> residual, bool save_z) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && w.scalar_type() == torch::kBFloat16);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && K % BK == 0);
  auto y = torch::empty({M, N}, x.options());
  torch::Tensor z;
  if (save_z) z = torch::empty({M, N}, x.options());
  c10::optional<torch::Tensor> bf;
  if (bias) bf = bias->contiguous().to(torch::kFloat32);
  int act_code = ACT_NONE;
  if (act == "gelu") act_code = ACT_GELU;
  else if (act == "gelu_tanh") act_code = ACT_GELU_TANH;
  else if (act == "quickgelu") act_code = ACT_QUICKGELU;
  else TORCH_CHECK(act.empty(), "unknown act ", act);

  const dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  const size_t shmem = 2 * (BM + BN) * PITCH * sizeof(short);
  auto stream = at::hip::getCurrentHIPStream();
  const bf16* resp = residual ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
  const float* biasp = bf ? bf->data_ptr<float>() : nullptr;
  bf16* zp = save_z ? reinterpret_cast<bf16*>(z.data_ptr()) : nullptr;

#define LAUNCH(ACTC, HB, HR, SP)                                                          \
  hipLaunchKernelGGL((gemm_nt_kernel<ACTC, HB, HR, SP>), grid, dim3(256), shmem, stream,  \
                     reinterpret_cast<const bf16*>(x.data_ptr()),                         \
                     reinterpret_cast<const bf16*>(w.data_ptr()), biasp, resp,            \
                     reinterpret_cast<bf16*>(y.data_ptr()), zp, M, N, K)
#define DISPATCH_ACT(HB, HR, SP)                                                          \
  switch (act_code) {                                                                     \
    case ACT_NONE: LAUNCH(ACT_NONE, HB, HR, SP); break;                                   \
    case ACT_GELU: LAUNCH(ACT_GELU, HB, HR, SP); break;                                   \
    case ACT_GELU_TANH: LAUNCH(ACT_GELU_TANH, HB, HR, SP); break;                         \
    case ACT_QUICKGELU: LAUNCH(ACT_QUICKGELU, HB, HR, SP); break;                         \
  }
  const bool hb = bias.has_value(), hr = residual.has_value();
  if (hb && hr && save_z) DISPATCH_ACT(true, true, true)
  else if (hb && hr) DISPATCH_ACT(true, true, false)
  else if (hb && save_z) DISPATCH_ACT(true, false, true)
  else if (hb) DISPATCH_ACT(true, false, false)
  else if (hr && save_z) DISPATCH_ACT(false, true, true)
  else if (hr) DISPATCH_ACT(false, true, false)
  else if (save_z) DISPATCH_ACT(false, false, true)
  else DISPATCH_ACT(false, false, false)
#undef DISPATCH_ACT
#undef LAUNCH
  if (save_z) return {y, z};
  return {y, torch::Tensor()};
}

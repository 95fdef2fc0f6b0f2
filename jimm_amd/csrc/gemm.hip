// K4/K6/K7/K8 — MFMA bf16 GEMM with fused bias/activation/residual epilogue.
//
// linear_fwd computes y = act(x @ w^T + bias) [+ residual] for x (M,K)
// row-major and w (N,K) row-major (torch Linear convention) — an "NT" GEMM
// on v_mfma_f32_16x16x32_bf16 tiles.
//
// Fast path (full 128x128 tiles, K%64==0 — every hot shape in the model zoo:
// ViT/CLIP/SigLIP widths are multiples of 128 and M = B*L*? lands on
// multiples of 128 for the bench batch sizes):
//   * 128x128 macro-tile, BK=64, 4 waves, 64x64 per wave (4x4 fragments);
//   * async global->LDS staging via __builtin_amdgcn_global_load_lds
//     width 16 (CDNA guide §5 ladder step 3: +69% over register staging);
//   * LDS image XOR-swizzled (chunk ^= row&7, 16B chunks) with the inverse
//     swizzle applied to the per-lane SOURCE address (guide rule 21) so the
//     ds_read_b128 fragment reads are bank-conflict-free;
//   * double-buffered LDS; stage(t+1) issued BEFORE compute(t) (minimum
//     2-phase recipe); __syncthreads() drains the in-flight DMA (vmcnt(0)
//     is emitted by the compiler with a glds outstanding);
//   * XCD-aware bijective blockIdx remap (guide T1) for L2 tile locality.
//
// Ragged shapes fall back to the generic guarded kernel below.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int PITCH = BK + 8;  // generic path only

// ---------------------------------------------------------------------------
// fast path: 128x128, glds staging, swizzled LDS
// ---------------------------------------------------------------------------

// LDS linear image: [128 rows][64 shorts] = 128 B/row, 16 KiB per operand
// tile; chunk c (16 B) of row r lives at byte r*128 + (c ^ (r&7))*16.

__device__ __forceinline__ void stage_tile_glds(const bf16* __restrict__ gsrc, int64_t ldg,
                                                short* lds_base, int tid) {
  // 256 threads stage 128x64 shorts (16 KiB): 4 rounds of 4 KiB; each wave
  // writes 1 KiB linearly at (round*4K + wave*1K); lane l covers offset
  // l*16. Source address carries the inverse swizzle.
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
#pragma unroll
  for (int round = 0; round < 4; ++round) {
    const int off = round * 4096 + wave * 1024 + lane * 16;  // byte offset in image
    const int row = off >> 7;              // /128
    const int chunk = (off >> 4) & 7;      // 16B chunk within row
    const int src_chunk = chunk ^ (row & 7);
    const bf16* g = gsrc + (int64_t)row * ldg + src_chunk * 8;
    // C-style casts perform the addrspace conversion (generic->AS1/AS3);
    // reinterpret_cast refuses (same idiom as CK's c_style_pointer_cast)
    typedef const __attribute__((address_space(1))) unsigned int* gp_t;
    typedef __attribute__((address_space(3))) unsigned int* lp_t;
    __builtin_amdgcn_global_load_lds((gp_t)(const void*)g,
                                     (lp_t)(void*)(reinterpret_cast<char*>(lds_base) + round * 4096 + wave * 1024),
                                     16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8_t lds_read_frag(const short* base, int row, int chunk) {
  const int byte = (row << 7) + ((chunk ^ (row & 7)) << 4);
  return *reinterpret_cast<const bf16x8_t*>(reinterpret_cast<const char*>(base) + byte);
}

template <int ACT, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE>
__global__ __launch_bounds__(256) void gemm_nt_fast_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W, const float* __restrict__ bias,
    const bf16* __restrict__ res, bf16* __restrict__ Y, bf16* __restrict__ Z,
    int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* smem_s = reinterpret_cast<short*>(smem);
  // buffers: x tiles at [0,8192) and [8192,16384); w tiles at +16384
  auto xs = [&](int buf) { return smem_s + buf * 8192; };
  auto ws = [&](int buf) { return smem_s + 16384 + buf * 8192; };

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  // XCD-aware bijective remap of the linear tile id (guide T1)
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
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;

  f32x4_t acc[4][4] = {};

  stage_tile_glds(X + (int64_t)m0 * K, K, xs(0), tid);
  stage_tile_glds(W + (int64_t)n0 * K, K, ws(0), tid);
  __syncthreads();

  const int nk = K / BK;
  for (int kt = 0; kt < nk; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nk) {
      stage_tile_glds(X + (int64_t)m0 * K + (kt + 1) * BK, K, xs(buf ^ 1), tid);
      stage_tile_glds(W + (int64_t)n0 * K + (kt + 1) * BK, K, ws(buf ^ 1), tid);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t xa[4], wb[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) xa[mi] = lds_read_frag(xs(buf), wm + 16 * mi + lo, 4 * ks + hi);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) wb[ni] = lds_read_frag(ws(buf), wn + 16 * ni + lo, 4 * ks + hi);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = MFMA16(xa[mi], wb[ni], acc[mi][ni]);
    }
    __syncthreads();  // also drains the in-flight global_load_lds (vmcnt 0)
  }

  // epilogue: C rows = m (hi*4+r of each 16-fragment), col = n (lo)
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
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

// ---------------------------------------------------------------------------
// generic (ragged) path — guarded staging, padded LDS, register staging
// ---------------------------------------------------------------------------

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

  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;

  f32x4_t acc[4][4] = {};

  auto stage = [&](int buf, int k0) {
    short* xd = xs + buf * BM * PITCH;
    short* wd = ws + buf * BN * PITCH;
    const int row = tid / 2;
    const int c0 = (tid & 1) * 32;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cc = c0 + h * 16;
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
    for (int ks = 0; ks < 2; ++ks) {
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

// 256x256-tile kernels: 8-phase (gemm8p.hip, the default — deep-pipelined
// counted-vmcnt schedule) and the older 2-phase (gemm256.hip, kept for A/B
// via JIMM_AMD_GEMM_TILE=256).
bool gemm256_supported(int64_t M, int64_t N, int64_t K);
void gemm_nt_256(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias_f32,
                 std::string act, c10::optional<torch::Tensor> residual, torch::Tensor y,
                 c10::optional<torch::Tensor> z);
bool gemm8p_supported(int64_t M, int64_t N, int64_t K);
void gemm_nt_8p(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias_f32,
                std::string act, c10::optional<torch::Tensor> residual, torch::Tensor y,
                c10::optional<torch::Tensor> z);

static int gemm_tile_mode() {
  // 0 = auto (8-phase when supported), 256 = force 2-phase 256, 128 = force
  // the 128-tile kernel (debug/A-B only)
  const char* env = getenv("JIMM_AMD_GEMM_TILE");
  if (!env) return 0;
  return atoi(env);
}

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

  const int tile_mode = gemm_tile_mode();
  if (tile_mode == 0 && gemm8p_supported(M, N, K)) {
    c10::optional<torch::Tensor> zopt;
    if (save_z) zopt = z;
    gemm_nt_8p(x, w, bf, act, residual, y, zopt);
    if (save_z) return {y, z};
    return {y, torch::Tensor()};
  }
  if (tile_mode != 128 && gemm256_supported(M, N, K)) {
    c10::optional<torch::Tensor> zopt;
    if (save_z) zopt = z;
    gemm_nt_256(x, w, bf, act, residual, y, zopt);
    if (save_z) return {y, z};
    return {y, torch::Tensor()};
  }

  auto stream = at::hip::getCurrentHIPStream();
  const bf16* resp = residual ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
  const float* biasp = bf ? bf->data_ptr<float>() : nullptr;
  bf16* zp = save_z ? reinterpret_cast<bf16*>(z.data_ptr()) : nullptr;
  const bool fast = (M % BM == 0) && (N % BN == 0);

#define LAUNCH(ACTC, HB, HR, SP)                                                          \
  do {                                                                                    \
    if (fast) {                                                                           \
      hipLaunchKernelGGL((gemm_nt_fast_kernel<ACTC, HB, HR, SP>),                         \
                         dim3((M / BM) * (N / BN)), dim3(256), 4 * 8192 * sizeof(short),  \
                         stream, reinterpret_cast<const bf16*>(x.data_ptr()),             \
                         reinterpret_cast<const bf16*>(w.data_ptr()), biasp, resp,        \
                         reinterpret_cast<bf16*>(y.data_ptr()), zp, M, N, K);             \
    } else {                                                                              \
      hipLaunchKernelGGL((gemm_nt_kernel<ACTC, HB, HR, SP>),                              \
                         dim3((M + BM - 1) / BM, (N + BN - 1) / BN), dim3(256),           \
                         2 * (BM + BN) * PITCH * sizeof(short), stream,                   \
                         reinterpret_cast<const bf16*>(x.data_ptr()),                     \
                         reinterpret_cast<const bf16*>(w.data_ptr()), biasp, resp,        \
                         reinterpret_cast<bf16*>(y.data_ptr()), zp, M, N, K);             \
    }                                                                                     \
  } while (0)
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

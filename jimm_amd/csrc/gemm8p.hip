// K4/K6/K7/K8 — 8-phase 256x256-tile MFMA bf16 GEMM with fused epilogue.
//
// Implements the CDNA4 guide's deep-pipelined 256-square structure
// (cdna_hip_programming.md §5 "The 256² 8-phase template"): the K-step is
// split into 4 sub-phases per K-tile, each {12x ds_read_b128 fragment loads |
// 1 half-tile global_load_lds prefetch | counted s_waitcnt vmcnt(4) | raw
// s_barrier | setprio(1) 16x mfma_f32_16x16x32_bf16 setprio(0) | raw
// s_barrier}.  The counted vmcnt (never 0 in the main loop) keeps 2 staged
// half-tiles in flight ACROSS barriers — the 2-phase vmcnt(0) structure in
// gemm256.hip drains the glds queue at every barrier, which is its ~900 TF
// structural ceiling; this schedule removes that drain (T3+T4, +28-41%, and
// enables T5 setprio, +21-25%).
//
// Geometry: 256x256 tile, BK=64, 512 threads = 8 waves.  Per phase p
// (quadrant (pr,pc) = (p>>1, p&1)) wave (wm in {0,1}, wn in 0..3) computes
// the 64x32 strip rows [pr*128+wm*64, +64) x cols [pc*128+wn*32, +32): the
// quadrant walk makes phases 0-1 consume only A-half0 / phases 2-3 A-half1
// (and B-half pc), so a half staged at phase f is first read 4 phases later
// and vmcnt(4) (= 2 half-tiles x 2 glds/wave in flight) certifies it.
//
// Stage schedule (tile t, phases p0..p3 stage): p0 -> B-half0(t+1),
// p1 -> B-half1(t+1), p2 -> A-half1(t+1), p3 -> A-half0(t+2).
// Slot reuse is safe: e.g. A-half0(t+2) lands in the buffer A[t&1][0] whose
// last reader was phase p1 of tile t (>= 2 barriers earlier).
//
// LDS: per operand 2 buffers x 2 half-images of [128 rows][64 shorts]
// (16 KiB each, XOR-swizzled chunk^=(row&7) with the inverse swizzle on the
// per-lane glds SOURCE address — §5.4 rule 21) = 128 KiB total.
//
// Epilogue fuses bias + gelu/gelu_tanh/quickgelu (+residual, +save-pre) like
// gemm256.hip.  M may be ragged (MGUARD clamps staging rows and predicates
// stores); N%256==0 and K%64==0, K>=128 required (gemm8p_supported).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTHREADS = 512;
constexpr int HALF_BYTES = 128 * BK * 2;  // 16 KiB per half-image

__device__ __forceinline__ void raw_barrier() {
  asm volatile("s_barrier" ::: "memory");
}

// LDS image swizzle (conflict-FREE for the fragment read pattern): the
// fragment reads put 16 consecutive rows (r0 multiple of 16) on the same
// 16-B chunk — with a row-major image rows r and r+8 always collide (same
// chunk xor, same row parity -> same 16-B slot of the 256-B bank row, 2-way
// on ds_read_b128).  Fix: interleave row PAIRS within one 256-B region and
// use row bit 3 to select the slot half:
//   slot(row, c) = (c ^ (row&7)) | (((row&1) ^ ((row>>3)&1)) << 3)
//   byte(row, c) = (row>>1)*256 + slot*16
// For rows r0..r0+15 at fixed c the 16 slots are pairwise distinct, so a
// wave's ds_read_b128 lane group touches 16 distinct 16-B slots = all 64
// banks (SQ_LDS_BANK_CONFLICT 0).  glds writes lane-linear, so the inverse
// permutation goes on the per-lane SOURCE address (§5.4 rule 21).
//
// Per-lane source base addresses (row/chunk fixed per lane; only the K
// offset advances) are precomputed once — the K-loop issues glds with
// base + kt*128 B, no per-phase 64-bit address rebuild.
//
// A half-image is staged by 512 threads x 16 B x 2 rounds; round r, thread
// tid covers image byte off = r*8192 + tid*16.
template <bool MGUARD, int SWZ>
__device__ __forceinline__ const bf16* stage_base(const bf16* __restrict__ gsrc,
                                                  int64_t ldg, int row0, int rows_total,
                                                  int round, int tid) {
  const int off = round * 8192 + tid * 16;
  int row, chunk;
  if constexpr (SWZ == 1) {
    // invert byte(row,c): pair = off/256, slot = (off/16)&15
    const int pair = off >> 8;
    const int slot = (off >> 4) & 15;
    const int half = slot >> 3;
    const int b = half ^ ((pair >> 2) & 1);  // row bit 3 = pair bit 2
    row = 2 * pair + b;
    chunk = (slot & 7) ^ (row & 7);
  } else {
    row = off >> 7;
    chunk = ((off >> 4) & 7) ^ (row & 7);
  }
  int grow = row0 + row;
  if (MGUARD) grow = grow < rows_total ? grow : rows_total - 1;
  return gsrc + (int64_t)grow * ldg + chunk * 8;
}

__device__ __forceinline__ void glds16(const bf16* g, char* lds_dst) {
  typedef const __attribute__((address_space(1))) unsigned int* gp_t;
  typedef __attribute__((address_space(3))) unsigned int* lp_t;
  __builtin_amdgcn_global_load_lds((gp_t)(const void*)g, (lp_t)(void*)lds_dst, 16, 0, 0);
}

__device__ __forceinline__ bf16x8_t lds_frag8p(const char* base, int byte_off) {
  return *reinterpret_cast<const bf16x8_t*>(base + byte_off);
}

template <int SWZ>
__device__ __forceinline__ int frag_off(int row, int chunk) {
  if constexpr (SWZ == 1) {
    const int slot = (chunk ^ (row & 7)) | ((((row & 1) ^ ((row >> 3) & 1))) << 3);
    return (row >> 1) * 256 + slot * 16;
  } else {
    return (row << 7) + ((chunk ^ (row & 7)) << 4);
  }
}

template <int ACT, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE, bool MGUARD, int SWZ = 1>
__global__ __launch_bounds__(NTHREADS, 2) void gemm_nt_8p_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W, const float* __restrict__ bias,
    const bf16* __restrict__ res, bf16* __restrict__ Y, bf16* __restrict__ Z,
    int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // A images: [buf][half] at (buf*2+half)*16KiB; B images at +64KiB
  auto As = [&](int buf, int h) { return smem + (buf * 2 + h) * HALF_BYTES; };
  auto Bs = [&](int buf, int h) { return smem + 4 * HALF_BYTES + (buf * 2 + h) * HALF_BYTES; };

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  const int mt = MGUARD ? (M + BM - 1) / BM : M / BM;
  const int nt = N / BN;
  const int nwg = mt * nt;
  int wg = blockIdx.x;
  {  // XCD-aware bijective remap (T1)
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int m0 = (wg / nt) * BM;
  const int n0 = (wg % nt) * BN;
  const int wm = (wave >> 2);      // 0..1 — M sub-position inside a quadrant
  const int wn = (wave & 3);       // 0..3 — N sub-position

  f32x4_t acc[4][4][2] = {};  // [phase][mi][ni]

  const int nk = K / BK;
  // Per-lane staging source bases (row/chunk fixed per lane; K advances by
  // 128 B per tile): [half][round] for each operand.
  const bf16* asrc[2][2];
  const bf16* bsrc[2][2];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int r = 0; r < 2; ++r) {
      asrc[h][r] = stage_base<MGUARD, SWZ>(X, K, m0 + h * 128, M, r, tid);
      bsrc[h][r] = stage_base<false, SWZ>(W + (int64_t)n0 * K, K, h * 128, N, r, tid);
    }
  const int ldst = (tid * 16) & 8191;        // lds dest byte of round 0
  auto stageA = [&](int kt, int h) {
    char* img = As(kt & 1, h);
    glds16(asrc[h][0] + (int64_t)kt * BK, img + ldst);
    glds16(asrc[h][1] + (int64_t)kt * BK, img + 8192 + ldst);
  };
  auto stageB = [&](int kt, int h) {
    char* img = Bs(kt & 1, h);
    glds16(bsrc[h][0] + (int64_t)kt * BK, img + ldst);
    glds16(bsrc[h][1] + (int64_t)kt * BK, img + 8192 + ldst);
  };

  // Per-lane fragment read byte offsets (loop-invariant; image base varies
  // by phase/buffer only).
  int offA[2][4], offB[2][2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) offA[ks][mi] = frag_off<SWZ>(wm * 64 + 16 * mi + lo, 4 * ks + hi);
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) offB[ks][ni] = frag_off<SWZ>(wn * 32 + 16 * ni + lo, 4 * ks + hi);
  }

  // Prologue: A0(0), B0(0), B1(0), A1(0), A0(1) then certify tile 0.
  stageA(0, 0);
  stageB(0, 0);
  stageB(0, 1);
  stageA(0, 1);
  if (nk > 1) {
    stageA(1, 0);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  raw_barrier();

  // ONE barrier per phase: a wave's phase-p+1 fragment reads may overlap
  // another wave's phase-p MFMA cluster (the LDS array drains under the
  // matrix pipe instead of strictly after it).  Safety: the slot a glds
  // overwrites was last read >= 2 barriers earlier, and a wave's reads
  // complete (hipcc's lgkmcnt) before its own MFMA -> before its next
  // barrier arrival, so no wave can see a slot mid-overwrite.
  for (int t = 0; t < nk; ++t) {
    const int buf = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int pr = p >> 1, pc = p & 1;
      // ---- fragment ds_reads for this phase (tile t images) ----
      bf16x8_t xa[2][4], wb[2][2];
      {
        const char* ai = As(buf, pr);
        const char* bi = Bs(buf, pc);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
          for (int mi = 0; mi < 4; ++mi) xa[ks][mi] = lds_frag8p(ai, offA[ks][mi]);
#pragma unroll
          for (int ni = 0; ni < 2; ++ni) wb[ks][ni] = lds_frag8p(bi, offB[ks][ni]);
        }
      }
      // ---- stage prefetch: p0->B0(t+1) p1->B1(t+1) p2->A1(t+1) p3->A0(t+2)
      if (p == 0 && t + 1 < nk) stageB(t + 1, 0);
      if (p == 1 && t + 1 < nk) stageB(t + 1, 1);
      if (p == 2 && t + 1 < nk) stageA(t + 1, 1);
      if (p == 3 && t + 2 < nk) stageA(t + 2, 0);
      // ---- counted wait: keep 2 half-tiles (4 loads/wave) in flight.
      // The last tile's first phase drains (its halves were the newest
      // stages and the stream has ended — vmcnt(4) would not cover them).
      if (t == nk - 1 && p == 0) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      }
      raw_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[p][mi][ni] = MFMA16(xa[ks][mi], wb[ks][ni], acc[p][mi][ni]);
      __builtin_amdgcn_s_setprio(0);
    }
  }

  // Epilogue: per phase strip rows pr*128+wm*64+16mi+hi*4+r, cols
  // pc*128+wn*32+16ni+lo (C layout: row = hi*4+r within a 16-fragment).
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int pr = p >> 1, pc = p & 1;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + pr * 128 + wm * 64 + 16 * mi + hi * 4 + r;
        if (MGUARD && m >= M) continue;
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const int n = n0 + pc * 128 + wn * 32 + 16 * ni + lo;
          float vpre = acc[p][mi][ni][r];
          if (HAS_BIAS) vpre += bias[n];
          if (SAVE_PRE) Z[(int64_t)m * N + n] = f2bf(vpre);
          float vy = act_fwd(vpre, ACT);
          if (HAS_RES) vy += bf2f(res[(int64_t)m * N + n]);
          Y[(int64_t)m * N + n] = f2bf(vy);
        }
      }
    }
  }
}

}  // namespace

static int gemm8p_swz() {
  const char* env = getenv("JIMM_AMD_GEMM_SWZ");
  return env ? atoi(env) : 1;
}

bool gemm8p_supported(int64_t M, int64_t N, int64_t K) {
  return M >= 1 && (N % BN == 0) && (K % BK == 0) && K >= 2 * BK;
}

void gemm_nt_8p(torch::Tensor x, torch::Tensor w, c10::optional<torch::Tensor> bias_f32,
                std::string act, c10::optional<torch::Tensor> residual, torch::Tensor y,
                c10::optional<torch::Tensor> z) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(gemm8p_supported(M, N, K));
  int act_code = ACT_NONE;
  if (act == "gelu") act_code = ACT_GELU;
  else if (act == "gelu_tanh") act_code = ACT_GELU_TANH;
  else if (act == "quickgelu") act_code = ACT_QUICKGELU;
  else TORCH_CHECK(act.empty(), "unknown act ", act);
  auto stream = at::hip::getCurrentHIPStream();
  const bf16* resp = residual ? reinterpret_cast<const bf16*>(residual->data_ptr()) : nullptr;
  const float* biasp = bias_f32 ? bias_f32->data_ptr<float>() : nullptr;
  bf16* zp = z ? reinterpret_cast<bf16*>(z->data_ptr()) : nullptr;
  const size_t shmem = 8 * HALF_BYTES;  // 128 KiB
  const bool mguard = (M % BM) != 0;
  const int mt = (M + BM - 1) / BM;

#define LAUNCH8P(ACTC, HB, HR, SP, MG)                                                     \
  do {                                                                                     \
    auto kfn = gemm8p_swz() == 1 ? gemm_nt_8p_kernel<ACTC, HB, HR, SP, MG, 1>          \
                                 : gemm_nt_8p_kernel<ACTC, HB, HR, SP, MG, 0>;                                    \
    static bool attr_set_##ACTC##HB##HR##SP##MG = [&] {                                    \
      hipFuncSetAttribute(reinterpret_cast<const void*>(kfn),                              \
                          hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);         \
      return true;                                                                         \
    }();                                                                                   \
    (void)attr_set_##ACTC##HB##HR##SP##MG;                                                 \
    hipLaunchKernelGGL(kfn, dim3(mt * (N / BN)), dim3(NTHREADS), shmem, stream,            \
                       reinterpret_cast<const bf16*>(x.data_ptr()),                        \
                       reinterpret_cast<const bf16*>(w.data_ptr()), biasp, resp,           \
                       reinterpret_cast<bf16*>(y.data_ptr()), zp, M, N, K);                \
  } while (0)
#define DISPATCH_MG(ACTC, HB, HR, SP)                                                      \
  do {                                                                                     \
    if (mguard) LAUNCH8P(ACTC, HB, HR, SP, true);                                          \
    else LAUNCH8P(ACTC, HB, HR, SP, false);                                                \
  } while (0)
#define DISPATCH_ACT8P(HB, HR, SP)                                                         \
  switch (act_code) {                                                                      \
    case ACT_NONE: DISPATCH_MG(ACT_NONE, HB, HR, SP); break;                               \
    case ACT_GELU: DISPATCH_MG(ACT_GELU, HB, HR, SP); break;                               \
    case ACT_GELU_TANH: DISPATCH_MG(ACT_GELU_TANH, HB, HR, SP); break;                     \
    case ACT_QUICKGELU: DISPATCH_MG(ACT_QUICKGELU, HB, HR, SP); break;                     \
  }
  const bool hb = bias_f32.has_value(), hr = residual.has_value(), sp = z.has_value();
  if (hb && hr && sp) DISPATCH_ACT8P(true, true, true)
  else if (hb && hr) DISPATCH_ACT8P(true, true, false)
  else if (hb && sp) DISPATCH_ACT8P(true, false, true)
  else if (hb) DISPATCH_ACT8P(true, false, false)
  else if (hr && sp) DISPATCH_ACT8P(false, true, true)
  else if (hr) DISPATCH_ACT8P(false, true, false)
  else if (sp) DISPATCH_ACT8P(false, false, true)
  else DISPATCH_ACT8P(false, false, false)
#undef DISPATCH_ACT8P
#undef DISPATCH_MG
#undef LAUNCH8P
}

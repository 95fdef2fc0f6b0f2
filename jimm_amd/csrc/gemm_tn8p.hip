// K15 — TN weight-gradient GEMM: dW[N,Kw] = dz^T @ X (contraction over the
// huge M = B*L dimension), 256x256 output tiles, split-M, MFMA bf16.
//
// Both operands are stored m-major ([M,N] / [M,Kw]) so both MFMA fragments
// need per-lane m-columns.  Staging: coalesced 16-B global loads into
// registers, then scattered ds_write_b128 into a transpose-friendly image;
// fragments come back by ds_read_b64_tr_b16 (the gfx950 hardware transpose
// read — no builtin, inline asm per guide §5.7 form (i): reads + their
// s_waitcnt lgkmcnt(0) in ONE statement, outputs "=&v").
//
// Image layout per operand per buffer (32 KiB, for one 64-m x 256-col tile):
//   byte(m, col) = kh*16384 + nt*1024 + qpos*128 + jm*32 + u*16
//   with kh = m>>5 (K-half), q' = (m>>2)&7, jm = m&3, qpos = (q'&1)*4+(q'>>1)
//   (even m-quads first, then odd), nt = col>>4, u = (col>>3)&1.
// A ds_read_b64_tr_b16 at per-lane address base + (lane&15)*2 + (lane>>4)*128
// delivers lane (lo,hi) the 4 elements m = 8*hi + (0..3) of column
// nt*16 + lo (the odd quads via immediate offset +512) — exactly the MFMA
// fragment m-octet per hi group, with the 32-lane groups touching disjoint
// bank rows (the verified T10 subtile pattern).
//
// Phase walk per 64-m tile mirrors gemm8p.hip: (kh, pr) x 4 phases, one
// __syncthreads() per phase (no glds in this kernel, so syncthreads does
// not drain the VMEM queue), X-side fragments reused in registers across
// the pr pair.  Staging for tile t+1: global loads issue at p0, dz-image
// ds_writes at p1, X-image writes at p2 (>= 2 barriers after the old
// buffer's last reader).  M-tail rows load zeros (no clamping — this is an
// accumulation).
//
// Split-M: grid.y walks m-chunks (slow index -> the chunk's operand rows
// stay L3-resident across its output tiles); partial results combine with
// fp32 atomicAdd into a zeroed C (one-shot store when grid.y == 1).
// Deterministic mode uses the rocBLAS path instead (ops/hip_linear.py).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef short bf16x4_t __attribute__((ext_vector_type(4)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define MFMA16(A, B, C) __builtin_amdgcn_mfma_f32_16x16x32_bf16(A, B, C, 0, 0, 0)

constexpr int BN = 256, BKW = 256, BMS = 64;  // n-tile, k-tile, m-step
constexpr int NTHREADS = 512;
constexpr int IMG_BYTES = 32768;  // one 64x256 bf16 image

__device__ __forceinline__ void raw_barrier_tn() {
  asm volatile("s_barrier" ::: "memory");
}

// image byte offset of element (m_local 0..63, col 0..255)
__device__ __forceinline__ int img_off(int m, int col) {
  const int kh = m >> 5, q = (m >> 2) & 7, jm = m & 3;
  const int qpos = (q & 1) * 4 + (q >> 1);
  return kh * 16384 + (col >> 4) * 1024 + qpos * 128 + jm * 32 + ((col >> 3) & 1) * 16;
}

// inverse: image byte offset o (16-B granular) -> (m_local, col)
__device__ __forceinline__ void img_inv(int o, int& m, int& col) {
  const int kh = o >> 14;
  const int nt = (o >> 10) & 15;
  const int qpos = (o >> 7) & 7;
  const int jm = (o >> 5) & 3;
  const int u = (o >> 4) & 1;
  const int q = ((qpos >> 2) & 1) + (qpos & 3) * 2;
  m = kh * 32 + q * 4 + jm;
  col = nt * 16 + u * 8;
}

__device__ __forceinline__ void glds16t(const bf16* g, char* lds_dst) {
  typedef const __attribute__((address_space(1))) unsigned int* gp_t;
  typedef __attribute__((address_space(3))) unsigned int* lp_t;
  __builtin_amdgcn_global_load_lds((gp_t)(const void*)g, (lp_t)(void*)lds_dst, 16, 0, 0);
}

// 4 tr reads of one frag-quad column block: returns the two m-octet halves
// of MFMA operands for fragments f and f+1?  We issue per-phase batches
// instead — see the asm blocks in the kernel.

template <bool ATOMIC, bool DBOUT = false, bool WSOUT = false>
__global__ __launch_bounds__(NTHREADS, 2) void gemm_tn_8p_kernel(
    const bf16* __restrict__ DZ, const bf16* __restrict__ X, float* __restrict__ C,
    int64_t M, int N, int Kw, int64_t chunk_m, float* __restrict__ DB) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // dz images: [buf]*32 KiB at 0; X images at 64 KiB.
  auto DZs = [&](int buf) { return smem + buf * IMG_BYTES; };
  auto Xs = [&](int buf) { return smem + 2 * IMG_BYTES + buf * IMG_BYTES; };

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wave = tid / WAVE;
  const int lo = lane & 15, hi = lane >> 4;

  const int nt_n = N / BN, nt_k = Kw / BKW;
  const int nwg = nt_n * nt_k;
  int wg = blockIdx.x;
  {  // XCD-aware bijective remap
    const int q = nwg / 8, r = nwg % 8;
    const int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int n0 = (wg / nt_k) * BN;
  const int k0 = (wg % nt_k) * BKW;
  const int wm = wave >> 2;  // 0..1 (n sub-position)
  const int wn = wave & 3;   // 0..3 (k sub-position)

  const int64_t m_begin = (int64_t)blockIdx.y * chunk_m;
  const int64_t m_end = m_begin + chunk_m < M ? m_begin + chunk_m : M;
  const int ntile = (int)((m_end - m_begin + BMS - 1) / BMS);
  if (ntile <= 0) return;

  f32x4_t acc[2][4][4] = {};  // [pr][mi][ni]
  // DBOUT: db[n] = sum_m dz[m][n] folded out of the A fragments already in
  // registers (the separate colsum kernel re-reads dz from HBM). Only the
  // k0==0 / wn==0 waves contribute (the same dz fragments are read by every
  // k-tile and every wn), so the whole-M coverage comes from the k0==0
  // column of workgroups across splits.
  float db_acc[2][4] = {};

  // Staging decode for this thread.  Fast path (full 64-row tiles): glds
  // with the image permutation on the per-lane SOURCE address — round r
  // covers image bytes [r*8192 + tid*16], whose content is the global chunk
  // img_inv() decodes.  Tail tile: coalesced register loads (zero-filled
  // out of range) + scattered ds_write_b128.
  int m_loc[4], col[4], io[4];     // register-staging (tail) decode
  int gm[4], gc[4];                // glds source decode per round
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int g = r * NTHREADS + tid;
    m_loc[r] = g >> 5;
    col[r] = (g & 31) * 8;
    io[r] = img_off(m_loc[r], col[r]);
    img_inv(r * 8192 + tid * 16, gm[r], gc[r]);
  }
  const int ldst16 = tid * 16;
  auto stage_glds = [&](int t, int buf) {
    const int64_t mt0 = m_begin + (int64_t)t * BMS;
    char* dzi = DZs(buf);
    char* xi = Xs(buf);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      glds16t(DZ + (mt0 + gm[r]) * N + n0 + gc[r], dzi + r * 8192 + ldst16);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      glds16t(X + (mt0 + gm[r]) * Kw + k0 + gc[r], xi + r * 8192 + ldst16);
  };

  typedef short s8 __attribute__((ext_vector_type(8)));
  // Tail-tile staging (coalesced guarded loads + scattered LDS writes;
  // round-by-round to keep register pressure low — the tail runs once).
  auto stage_reg_tail = [&](int t, int buf) {
    const int64_t mt0 = m_begin + (int64_t)t * BMS;
    char* dzi = DZs(buf);
    char* xi = Xs(buf);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int64_t m = mt0 + m_loc[r];
      s8 a = s8{}, b = s8{};
      if (m < m_end) {
        a = *reinterpret_cast<const s8*>(DZ + m * N + n0 + col[r]);
        b = *reinterpret_cast<const s8*>(X + m * Kw + k0 + col[r]);
      }
      *reinterpret_cast<s8*>(dzi + io[r]) = a;
      *reinterpret_cast<s8*>(xi + io[r]) = b;
    }
  };

  // Per-lane tr-read address term: lane*8 B — each lane points at its own
  // consecutive b64 slot; the hardware transposes within each 16-lane group
  // (verified by ext.tr16_probe mode 1: lane l elem j reads short
  // (l>>4)*64 + (l&15) + j*16 relative to lane 0's address).
  const int lane_term = lane * 8;

  // prologue: stage tile 0 (glds when the tile has all 64 rows)
  const bool tile0_full = m_begin + BMS <= m_end;
  if (tile0_full) {
    stage_glds(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else {
    stage_reg_tail(0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  raw_barrier_tn();

  for (int t = 0; t < ntile; ++t) {
    const int buf = t & 1;
    const bool more = t + 1 < ntile;
    const bool next_full = m_begin + (int64_t)(t + 2) * BMS <= m_end;
    bf16x8_t xb[4];  // X-side fragments, reused across the pr pair
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      // ---------- phase (kh, pr=0) ----------
      {
        // staging for t+1: full tiles go by glds at the kh=0 pr=1 phase;
        // the ragged tail tile goes by registers (loads here, writes in
        // the kh=1 phases)
        if (kh == 1 && more && !next_full) stage_reg_tail(t + 1, buf ^ 1);
        // B (X image) frags ni=0..3 at nt = wn*4+ni, plus A (dz) frags
        // mi=0..3 at nt = wm*4+mi (pr=0).  8+8 tr reads + lgkmcnt(0) in one
        // asm (form i).
        typedef const __attribute__((address_space(3))) char* lds_p;
        const lds_p bxi = (lds_p)(const void*)(Xs(buf) + kh * 16384 + wn * 4096 + lane_term);
        const lds_p adi = (lds_p)(const void*)(DZs(buf) + kh * 16384 + wm * 4096 + lane_term);
        bf16x4_t b0l, b0h, b1l, b1h, b2l, b2h, b3l, b3h;
        bf16x4_t a0l, a0h, a1l, a1h, a2l, a2h, a3l, a3h;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %16 offset:0\n\t"
            "ds_read_b64_tr_b16 %1, %16 offset:512\n\t"
            "ds_read_b64_tr_b16 %2, %16 offset:1024\n\t"
            "ds_read_b64_tr_b16 %3, %16 offset:1536\n\t"
            "ds_read_b64_tr_b16 %4, %16 offset:2048\n\t"
            "ds_read_b64_tr_b16 %5, %16 offset:2560\n\t"
            "ds_read_b64_tr_b16 %6, %16 offset:3072\n\t"
            "ds_read_b64_tr_b16 %7, %16 offset:3584\n\t"
            "ds_read_b64_tr_b16 %8, %17 offset:0\n\t"
            "ds_read_b64_tr_b16 %9, %17 offset:512\n\t"
            "ds_read_b64_tr_b16 %10, %17 offset:1024\n\t"
            "ds_read_b64_tr_b16 %11, %17 offset:1536\n\t"
            "ds_read_b64_tr_b16 %12, %17 offset:2048\n\t"
            "ds_read_b64_tr_b16 %13, %17 offset:2560\n\t"
            "ds_read_b64_tr_b16 %14, %17 offset:3072\n\t"
            "ds_read_b64_tr_b16 %15, %17 offset:3584\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(b0l), "=&v"(b0h), "=&v"(b1l), "=&v"(b1h), "=&v"(b2l), "=&v"(b2h),
              "=&v"(b3l), "=&v"(b3h), "=&v"(a0l), "=&v"(a0h), "=&v"(a1l), "=&v"(a1h),
              "=&v"(a2l), "=&v"(a2h), "=&v"(a3l), "=&v"(a3h)
            : "v"(bxi), "v"(adi)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
        xb[0] = __builtin_shufflevector(b0l, b0h, 0, 1, 2, 3, 4, 5, 6, 7);
        xb[1] = __builtin_shufflevector(b1l, b1h, 0, 1, 2, 3, 4, 5, 6, 7);
        xb[2] = __builtin_shufflevector(b2l, b2h, 0, 1, 2, 3, 4, 5, 6, 7);
        xb[3] = __builtin_shufflevector(b3l, b3h, 0, 1, 2, 3, 4, 5, 6, 7);
        bf16x8_t ad[4];
        ad[0] = __builtin_shufflevector(a0l, a0h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[1] = __builtin_shufflevector(a1l, a1h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[2] = __builtin_shufflevector(a2l, a2h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[3] = __builtin_shufflevector(a3l, a3h, 0, 1, 2, 3, 4, 5, 6, 7);
        if (DBOUT && wn == 0 && k0 == 0) {
#pragma unroll
          for (int mi = 0; mi < 4; ++mi) {
            const bf16* av = reinterpret_cast<const bf16*>(&ad[mi]);
            float s8 = 0.f;
#pragma unroll
            for (int j = 0; j < 8; ++j) s8 += bf2f(av[j]);
            db_acc[0][mi] += s8;
          }
        }
        raw_barrier_tn();
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[0][mi][ni] = MFMA16(ad[mi], xb[ni], acc[0][mi][ni]);
        __builtin_amdgcn_s_setprio(0);
      }
      // ---------- phase (kh, pr=1) ----------
      {
        if (kh == 0 && more && next_full) stage_glds(t + 1, buf ^ 1);
        typedef const __attribute__((address_space(3))) char* lds_p;
        const lds_p adi = (lds_p)(const void*)(DZs(buf) + kh * 16384 + 8192 + wm * 4096 + lane_term);
        bf16x4_t a0l, a0h, a1l, a1h, a2l, a2h, a3l, a3h;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %8 offset:0\n\t"
            "ds_read_b64_tr_b16 %1, %8 offset:512\n\t"
            "ds_read_b64_tr_b16 %2, %8 offset:1024\n\t"
            "ds_read_b64_tr_b16 %3, %8 offset:1536\n\t"
            "ds_read_b64_tr_b16 %4, %8 offset:2048\n\t"
            "ds_read_b64_tr_b16 %5, %8 offset:2560\n\t"
            "ds_read_b64_tr_b16 %6, %8 offset:3072\n\t"
            "ds_read_b64_tr_b16 %7, %8 offset:3584\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(a0l), "=&v"(a0h), "=&v"(a1l), "=&v"(a1h), "=&v"(a2l), "=&v"(a2h),
              "=&v"(a3l), "=&v"(a3h)
            : "v"(adi)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
        bf16x8_t ad[4];
        ad[0] = __builtin_shufflevector(a0l, a0h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[1] = __builtin_shufflevector(a1l, a1h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[2] = __builtin_shufflevector(a2l, a2h, 0, 1, 2, 3, 4, 5, 6, 7);
        ad[3] = __builtin_shufflevector(a3l, a3h, 0, 1, 2, 3, 4, 5, 6, 7);
        if (DBOUT && wn == 0 && k0 == 0) {
#pragma unroll
          for (int mi = 0; mi < 4; ++mi) {
            const bf16* av = reinterpret_cast<const bf16*>(&ad[mi]);
            float s8 = 0.f;
#pragma unroll
            for (int j = 0; j < 8; ++j) s8 += bf2f(av[j]);
            db_acc[1][mi] += s8;
          }
        }
        // kh==1 (phase p3): drain the glds staged at p1 before the barrier
        // that gates the next tile's reads
        if (kh == 1 && more && next_full) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        raw_barrier_tn();
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            acc[1][mi][ni] = MFMA16(ad[mi], xb[ni], acc[1][mi][ni]);
        __builtin_amdgcn_s_setprio(0);
      }
    }
  }

  if (DBOUT && wn == 0 && k0 == 0) {
    // A-frag row for lane = lo: db_acc[pr][mi] covers n = n0 + pr*128 +
    // wm*64 + 16*mi + lo over this wave's m chunks (hi groups); reduce
    // over hi, one atomic per n per split
#pragma unroll
    for (int pr = 0; pr < 2; ++pr)
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        float v_ = db_acc[pr][mi];
        v_ += __shfl_xor(v_, 16, WAVE);
        v_ += __shfl_xor(v_, 32, WAVE);
        if (hi == 0) atomicAdd(DB + n0 + pr * 128 + wm * 64 + 16 * mi + lo, v_);
      }
  }

  // Epilogue: acc[pr][mi][ni] -> C rows n = n0 + pr*128 + wm*64 + 16mi +
  // hi*4 + r, cols k = k0 + wn*64 + 16ni + lo.
#pragma unroll
  for (int pr = 0; pr < 2; ++pr)
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = n0 + pr * 128 + wm * 64 + 16 * mi + hi * 4 + r;
#pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
          const int k = k0 + wn * 64 + 16 * ni + lo;
          // WSOUT: each split owns a private C slice (no atomics; a reduce
          // kernel folds the splits) — A/B alternative to the atomic
          // combine, selected by JIMM_AMD_DW_WS=1
          float* dst = C + (WSOUT ? (int64_t)blockIdx.y * N * Kw : 0) +
                       (int64_t)n * Kw + k;
          if (ATOMIC && !WSOUT) atomicAdd(dst, acc[pr][mi][ni][r]);
          else *dst = acc[pr][mi][ni][r];
        }
      }
}

__global__ void ws_reduce_kernel(const float* __restrict__ ws, float* __restrict__ out,
                                 int64_t n, int splits) {
  typedef float f32x4_r __attribute__((ext_vector_type(4)));
  const int64_t nv = n / 4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    f32x4_r acc = *reinterpret_cast<const f32x4_r*>(ws + i * 4);
    for (int sp = 1; sp < splits; ++sp) {
      const f32x4_r v = *reinterpret_cast<const f32x4_r*>(ws + (int64_t)sp * n + i * 4);
      acc += v;
    }
    *reinterpret_cast<f32x4_r*>(out + i * 4) = acc;
  }
}

}  // namespace

bool gemm_tn8p_supported(int64_t M, int64_t N, int64_t K) {
  return M >= 1 && (N % BN == 0) && (K % BKW == 0);
}

static torch::Tensor tn8p_run(torch::Tensor dz, torch::Tensor x, float* dbp) {
  // dz (M, N) bf16, x (M, Kw) bf16 -> dW (N, Kw) fp32 = dz^T @ x
  // (dbp != nullptr: also fold db[n] = colsum(dz) out of the A fragments)
  TORCH_CHECK(dz.is_cuda() && dz.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dz.scalar_type() == torch::kBFloat16 && x.scalar_type() == torch::kBFloat16);
  const int64_t M = dz.size(0);
  const int N = dz.size(1), Kw = x.size(1);
  TORCH_CHECK(x.size(0) == M && gemm_tn8p_supported(M, N, Kw));
  auto stream = at::hip::getCurrentHIPStream();
  const int tiles = (N / BN) * (Kw / BKW);
  // split-M so total blocks ~ 2-3 waves of the 256 CUs (JIMM_AMD_DW_SPLITM
  // overrides for tuning; higher split = more parallelism but splitm x
  // C-sized fp32 atomic traffic)
  static const int splitm_env = [] {
    const char* e = getenv("JIMM_AMD_DW_SPLITM");
    return e ? atoi(e) : 0;
  }();
  // r02 sweep at M=128*577: small-tile shapes (proj 1024x1024 -> 16 tiles)
  // are atomic-bound and want ~208 total WGs (430 -> 650 TF/s); larger
  // shapes peak around <= 20 splits (qkv 48 tiles: 27 -> 20 gave +6%).
  int heur = tiles <= 16 ? (208 + tiles - 1) / tiles
                         : (int)std::min<int64_t>((1280 + tiles - 1) / tiles, 20);
  int splitm = splitm_env > 0
                   ? splitm_env
                   : (int)std::min<int64_t>(heur, (M + BMS - 1) / BMS);
  if ((int64_t)splitm > (M + BMS - 1) / BMS) splitm = (int)((M + BMS - 1) / BMS);
  if (splitm < 1) splitm = 1;
  int64_t chunk = ((M + splitm - 1) / splitm + BMS - 1) / BMS * BMS;
  splitm = (int)((M + chunk - 1) / chunk);
  // Workspace combine (default): each split writes a private C slice and a
  // vectorized reduce folds them — measured 662-745 -> 684-824 TF/s over
  // the fp32 atomic combine on the block shapes, and the fixed-order
  // reduce makes the whole dW deterministic. JIMM_AMD_DW_WS=0 restores
  // the atomic combine for A/B (nondeterministic accumulation order).
  static const bool ws_mode = [] {
    const char* e = getenv("JIMM_AMD_DW_WS");
    return !(e && e[0] == '0');
  }();
  const bool use_ws = ws_mode && splitm > 1;
  auto C = (splitm > 1 && !use_ws)
               ? torch::zeros({(int64_t)N, (int64_t)Kw}, dz.options().dtype(torch::kFloat32))
               : torch::empty({(int64_t)N, (int64_t)Kw}, dz.options().dtype(torch::kFloat32));
  torch::Tensor ws;
  if (use_ws)
    ws = torch::empty({(int64_t)splitm, (int64_t)N, (int64_t)Kw},
                      dz.options().dtype(torch::kFloat32));
  const size_t shmem = 4 * IMG_BYTES;  // 128 KiB
#define LAUNCH_TN(AT, DBO, WSO)                                                            \
  do {                                                                                     \
    auto kfn = gemm_tn_8p_kernel<AT, DBO, WSO>;                                            \
    static bool attr_##AT##DBO##WSO = [&] {                                                \
      hipFuncSetAttribute(reinterpret_cast<const void*>(kfn),                              \
                          hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);         \
      return true;                                                                         \
    }();                                                                                   \
    (void)attr_##AT##DBO##WSO;                                                             \
    hipLaunchKernelGGL(kfn, dim3(tiles, splitm), dim3(NTHREADS), shmem, stream,            \
                       reinterpret_cast<const bf16*>(dz.data_ptr()),                       \
                       reinterpret_cast<const bf16*>(x.data_ptr()),                        \
                       WSO ? ws.data_ptr<float>() : C.data_ptr<float>(), M, N, Kw, chunk,  \
                       dbp);                                                               \
  } while (0)
  if (use_ws) {
    if (dbp) LAUNCH_TN(false, true, true);
    else LAUNCH_TN(false, false, true);
    const int64_t n = (int64_t)N * Kw;
    const int block = 256;
    const int grid = (int)std::min<int64_t>((n / 4 + block - 1) / block, 8192);
    hipLaunchKernelGGL(ws_reduce_kernel, dim3(grid), dim3(block), 0, stream,
                       ws.data_ptr<float>(), C.data_ptr<float>(), n, splitm);
  } else if (dbp && splitm > 1) LAUNCH_TN(true, true, false);
  else if (dbp) LAUNCH_TN(false, true, false);
  else if (splitm > 1) LAUNCH_TN(true, false, false);
  else LAUNCH_TN(false, false, false);
#undef LAUNCH_TN
  return C;
}

torch::Tensor gemm_tn_8p(torch::Tensor dz, torch::Tensor x) {
  return tn8p_run(dz, x, nullptr);
}

std::vector<torch::Tensor> gemm_tn_8p_db(torch::Tensor dz, torch::Tensor x) {
  // fused dW + bias-grad: {dW (N, Kw) fp32, db (N) fp32}
  auto db = torch::zeros({dz.size(1)}, dz.options().dtype(torch::kFloat32));
  auto C = tn8p_run(dz, x, db.data_ptr<float>());
  return {C, db};
}

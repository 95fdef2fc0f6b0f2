// K15 — attention backward helper kernels (fused elementwise pieces of the
// recompute-P composite; the GEMMs run through rocBLAS).
//
//   attn_bwd_p : S -> P = exp(scale*S - lse[row]) in place (bf16, causal opt)
//   attn_d     : D[row] = sum_d dO[row,d] * O[row,d]   (fp32)
//   attn_ds    : dP -> dS = P * (dP - D[row]) * scale  in place (bf16)
//
// These replace a chain of eager fp32 casts/exps/muls over (B,H,Lq,Lk)
// tensors (~30% of step time in the first profile, profiles/r01_*).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short short8_t __attribute__((ext_vector_type(8)));

template <bool CAUSAL>
__global__ void attn_bwd_p_kernel(bf16* __restrict__ s, const float* __restrict__ lse,
                                  int64_t nrows, int Lk, float scale) {
  // one wave per row; rows = B*H*Lq, row index into lse directly
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t row = (int64_t)blockIdx.x * wpb + wave; row < nrows;
       row += (int64_t)gridDim.x * wpb) {
    const float l = lse[row];
    const int qi = CAUSAL ? (int)(row % /*Lq==Lk for causal*/ Lk) : 0;
    bf16* sr = s + row * Lk;
    // rows are generally NOT 16B-aligned (Lk=197 etc.): scalar head until
    // aligned, vectorized middle, scalar tail
    const int head = min((int)(((16 - ((uintptr_t)sr & 15)) & 15) / 2), Lk);
    const int Lv = head + ((Lk - head) & ~7);
    for (int c = lane; c < head; c += WAVE) {
      float p = __expf(bf2f(sr[c]) * scale - l);
      if (CAUSAL && c > qi) p = 0.f;
      sr[c] = f2bf(p);
    }
    for (int c = head + lane * 8; c < Lv; c += WAVE * 8) {
      short8_t v = *reinterpret_cast<const short8_t*>(sr + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(bfs2f(v[j]) * scale - l);
        if (CAUSAL && c + j > qi) p = 0.f;
        v[j] = f2bfs(p);
      }
      *reinterpret_cast<short8_t*>(sr + c) = v;
    }
    for (int c = Lv + lane; c < Lk; c += WAVE) {
      float p = __expf(bf2f(sr[c]) * scale - l);
      if (CAUSAL && c > qi) p = 0.f;
      sr[c] = f2bf(p);
    }
  }
}

__global__ void attn_d_kernel(const bf16* __restrict__ dO, const bf16* __restrict__ O,
                              float* __restrict__ D, int64_t nrows, int d) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t row = (int64_t)blockIdx.x * wpb + wave; row < nrows;
       row += (int64_t)gridDim.x * wpb) {
    float acc = 0.f;
    const bf16* a = dO + row * d;
    const bf16* b = O + row * d;
    for (int c = lane; c < d; c += WAVE) acc += bf2f(a[c]) * bf2f(b[c]);
    acc = wave_reduce_sum(acc);
    if (lane == 0) D[row] = acc;
  }
}

__global__ void attn_ds_kernel(bf16* __restrict__ dp, const bf16* __restrict__ p,
                               const float* __restrict__ D, int64_t nrows, int Lk,
                               float scale) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t row = (int64_t)blockIdx.x * wpb + wave; row < nrows;
       row += (int64_t)gridDim.x * wpb) {
    const float dv = D[row];
    bf16* dpr = dp + row * Lk;
    const bf16* pr = p + row * Lk;
    const int head = min((int)(((16 - ((uintptr_t)dpr & 15)) & 15) / 2), Lk);
    const int Lv = head + ((Lk - head) & ~7);
    for (int c = lane; c < head; c += WAVE)
      dpr[c] = f2bf(bf2f(pr[c]) * (bf2f(dpr[c]) - dv) * scale);
    for (int c = head + lane * 8; c < Lv; c += WAVE * 8) {
      short8_t a = *reinterpret_cast<const short8_t*>(dpr + c);
      short8_t b = *reinterpret_cast<const short8_t*>(pr + c);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        a[j] = f2bfs(bfs2f(b[j]) * (bfs2f(a[j]) - dv) * scale);
      *reinterpret_cast<short8_t*>(dpr + c) = a;
    }
    for (int c = Lv + lane; c < Lk; c += WAVE)
      dpr[c] = f2bf(bf2f(pr[c]) * (bf2f(dpr[c]) - dv) * scale);
  }
}

}  // namespace

void attn_bwd_p(torch::Tensor s, torch::Tensor lse, bool causal, double scale) {
  TORCH_CHECK(s.is_cuda() && s.is_contiguous() && s.scalar_type() == torch::kBFloat16);
  const int Lk = s.size(-1);
  const int64_t nrows = s.numel() / Lk;
  if (causal) TORCH_CHECK(s.size(-2) == Lk, "causal requires Lq==Lk");
  const int grid = (int)std::min<int64_t>((nrows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  if (causal)
    hipLaunchKernelGGL((attn_bwd_p_kernel<true>), dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<bf16*>(s.data_ptr()), lse.data_ptr<float>(), nrows,
                       Lk, (float)scale);
  else
    hipLaunchKernelGGL((attn_bwd_p_kernel<false>), dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<bf16*>(s.data_ptr()), lse.data_ptr<float>(), nrows,
                       Lk, (float)scale);
}

torch::Tensor attn_d(torch::Tensor dO, torch::Tensor O) {
  TORCH_CHECK(dO.is_cuda() && dO.is_contiguous() && O.is_contiguous());
  const int d = dO.size(-1);
  const int64_t nrows = dO.numel() / d;
  auto D = torch::empty({nrows}, dO.options().dtype(torch::kFloat32));
  const int grid = (int)std::min<int64_t>((nrows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_d_kernel, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const bf16*>(dO.data_ptr()),
                     reinterpret_cast<const bf16*>(O.data_ptr()), D.data_ptr<float>(),
                     nrows, d);
  return D;
}

void attn_ds(torch::Tensor dp, torch::Tensor p, torch::Tensor D, double scale) {
  TORCH_CHECK(dp.is_cuda() && dp.is_contiguous() && p.is_contiguous());
  const int Lk = dp.size(-1);
  const int64_t nrows = dp.numel() / Lk;
  const int grid = (int)std::min<int64_t>((nrows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_ds_kernel, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<bf16*>(dp.data_ptr()),
                     reinterpret_cast<const bf16*>(p.data_ptr()), D.data_ptr<float>(),
                     nrows, Lk, (float)scale);
}

// Fused elementwise kernels:
//   * bias_act_fwd  — z += bias (in place, kept as pre-activation for bwd),
//                     y = act(z) [+ residual]; acts: gelu / gelu_tanh /
//                     quickgelu (K7 epilogue; transformer.py:92-103)
//   * act_bwd       — dz = dy * act'(z)
//   * im2col/col2im — K1 patch-embed gather for kernel==stride convs
//                     (common/vit.py:153-165,228-230): pure permute-gather,
//                     coalesced on the cols side.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_fp8.h>

#include "common.h"

namespace {

constexpr int kMaxGrid = 2048;

template <typename T, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE>
__global__ void bias_act_kernel(T* __restrict__ z, const T* __restrict__ bias,
                                const T* __restrict__ res, T* __restrict__ y, int64_t n,
                                int M, int act) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = (float)z[i];
    if (HAS_BIAS) v += (float)bias[i % M];
    if (SAVE_PRE) z[i] = (T)v;  // pre-activation saved for act_bwd
    float o = act_fwd(v, act);
    if (HAS_RES) o += (float)res[i];
    y[i] = (T)o;
  }
}

template <typename T, bool HAS_BIAS, bool HAS_RES, bool SAVE_PRE, bool FP8O = false>
__global__ void bias_act_vec_kernel(T* __restrict__ z, const T* __restrict__ bias,
                                    const T* __restrict__ res, T* __restrict__ y,
                                    int64_t nv, int Mv, int act,
                                    unsigned char* __restrict__ y8 = nullptr,
                                    const float* __restrict__ scale8 = nullptr,
                                    unsigned int* __restrict__ amax_bits = nullptr) {
  // vectorized x8 variant; requires M % 8 == 0 (Mv = M/8, nv = n/8).
  // FP8O: also emit the delayed-scaled e4m3 copy of y (see layernorm.hip).
  constexpr int V = 8;
  float rs8 = 1.f, tmax = 0.f;
  if (FP8O) rs8 = 1.f / scale8[0];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    float zv[V], o[V];
    vload_f32<V>(z + i * V, zv);
    if (HAS_BIAS) {
      float bv[V];
      vload_f32<V>(bias + (i % Mv) * V, bv);
#pragma unroll
      for (int j = 0; j < V; ++j) zv[j] += bv[j];
    }
    if (SAVE_PRE) vstore_f32<V>(z + i * V, zv);
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = act_fwd(zv[j], act);
    if (HAS_RES) {
      float rv[V];
      vload_f32<V>(res + i * V, rv);
#pragma unroll
      for (int j = 0; j < V; ++j) o[j] += rv[j];
    }
    vstore_f32<V>(y + i * V, o);
    if (FP8O) {
      unsigned short q8[V / 2];
#pragma unroll
      for (int j = 0; j < V; ++j) tmax = fmaxf(tmax, fabsf(o[j]));
#pragma unroll
      for (int j = 0; j < V; j += 2) q8[j / 2] = cvt2_e4m3(o[j] * rs8, o[j + 1] * rs8);
      *reinterpret_cast<uint2*>(y8 + i * V) = *reinterpret_cast<uint2*>(q8);
    }
  }
  if (FP8O) {
    tmax = wave_reduce_max(tmax);
    if ((threadIdx.x % WAVE) == 0) atomicMax(amax_bits, __float_as_uint(tmax));
  }
}

template <typename T>
__global__ void act_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ z,
                               T* __restrict__ dz, int64_t n, int act) {
  constexpr int V = 8;
  const int64_t nv = n / V;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    float dyv[V], zv[V], o[V];
    vload_f32<V>(dy + i * V, dyv);
    vload_f32<V>(z + i * V, zv);
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = dyv[j] * act_grad(zv[j], act);
    vstore_f32<V>(dz + i * V, o);
  }
  // scalar tail
  for (int64_t i = nv * V + (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dz[i] = (T)((float)dy[i] * act_grad((float)z[i], act));
}

// db[col] = sum_rows dz[row][col] — bias gradient (K15). Thread owns 8
// consecutive columns (b128 loads), 4-deep unrolled row loop keeps loads in
// flight; partial sums land in a (gy*rstep, N) fp32 workspace (NO global
// atomics — v1/v2 with atomics measured 133/262 us for (50k,768); the
// per-address serialization dominated). Host reduces the workspace.
template <typename T>
__global__ void colsum_kernel(const T* __restrict__ dz, float* __restrict__ ws, int64_t M,
                              int N, int rows_per_wg) {
  constexpr int V = 8;
  const int strips = N / V;                       // column strips of 8
  const int strip = (blockIdx.x * blockDim.x + threadIdx.x) % strips;
  const int rlane = (blockIdx.x * blockDim.x + threadIdx.x) / strips;
  const int rstep = (gridDim.x * blockDim.x) / strips;  // row-parallel threads per strip
  if (rlane >= rstep) return;  // leftover threads would double-count rows
  const int c0 = strip * V;
  const int64_t r0 = (int64_t)blockIdx.y * rows_per_wg;
  int64_t r1 = r0 + rows_per_wg;
  if (r1 > M) r1 = M;
  float acc[V] = {};
  int64_t r = r0 + rlane;
  for (; r + 3 * rstep < r1; r += 4 * rstep) {
    float v0[V], v1[V], v2[V], v3[V];
    vload_f32<V>(dz + r * N + c0, v0);
    vload_f32<V>(dz + (r + rstep) * N + c0, v1);
    vload_f32<V>(dz + (r + 2 * rstep) * N + c0, v2);
    vload_f32<V>(dz + (r + 3 * rstep) * N + c0, v3);
#pragma unroll
    for (int j = 0; j < V; ++j) acc[j] += (v0[j] + v1[j]) + (v2[j] + v3[j]);
  }
  for (; r < r1; r += rstep) {
    float v[V];
    vload_f32<V>(dz + r * N + c0, v);
#pragma unroll
    for (int j = 0; j < V; ++j) acc[j] += v[j];
  }
  float* out = ws + ((int64_t)blockIdx.y * rstep + rlane) * N + c0;
#pragma unroll
  for (int j = 0; j < V; ++j) out[j] = acc[j];
}

// cols[b*h*w + ph*w + pw][c*P*P + i*P + j] = img[b][c][ph*P+i][pw*P+j]
template <typename T>
__global__ void im2col_kernel(const T* __restrict__ img, T* __restrict__ cols, int B,
                              int C, int HH, int WW, int P) {
  const int h = HH / P, w = WW / P;
  const int64_t ncols = (int64_t)C * P * P;
  const int64_t total = (int64_t)B * h * w * ncols;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / ncols;
    const int col = (int)(idx - row * ncols);
    const int pw = (int)(row % w);
    const int ph = (int)((row / w) % h);
    const int b = (int)(row / ((int64_t)h * w));
    const int j = col % P;
    const int i = (col / P) % P;
    const int c = col / (P * P);
    cols[idx] = img[(((int64_t)b * C + c) * HH + ph * P + i) * WW + pw * P + j];
  }
}

template <typename T>
__global__ void col2im_kernel(const T* __restrict__ cols, T* __restrict__ img, int B,
                              int C, int HH, int WW, int P) {
  const int h = HH / P, w = WW / P;
  const int64_t total = (int64_t)B * C * HH * WW;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int x = (int)(idx % WW);
    const int y = (int)((idx / WW) % HH);
    const int c = (int)((idx / ((int64_t)WW * HH)) % C);
    const int b = (int)(idx / ((int64_t)WW * HH * C));
    const int ph = y / P, i = y % P;
    const int pw = x / P, j = x % P;
    const int64_t row = ((int64_t)b * h + ph) * w + pw;
    const int64_t col = ((int64_t)c * P + i) * P + j;
    img[idx] = cols[row * (int64_t)C * P * P + col];
  }
}

// K2: y[b,0,:] = cls + pos[0]; y[b,1+l,:] = x[b,l,:] + pos[1+l,:]
// (CLS-token concat + position-embedding add in one pass; without cls it is
// a plain broadcast pos add). Reference: common/vit.py:232-241.
template <typename T, bool HAS_CLS>
__global__ void cls_pos_kernel(const T* __restrict__ x, const T* __restrict__ cls,
                               const T* __restrict__ pos, T* __restrict__ y, int B,
                               int Lout, int Hv) {
  // Hv = H/8 vector groups; one thread handles one 8-group
  constexpr int V = 8;
  const int64_t total = (int64_t)B * Lout * Hv;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int hv = (int)(idx % Hv);
    const int l = (int)((idx / Hv) % Lout);
    const int64_t b = idx / ((int64_t)Hv * Lout);
    float pv[V], sv[V], o[V];
    vload_f32<V>(pos + ((int64_t)l * Hv + hv) * V, pv);
    if (HAS_CLS && l == 0) {
      vload_f32<V>(cls + (int64_t)hv * V, sv);
    } else {
      const int lx = HAS_CLS ? l - 1 : l;
      vload_f32<V>(x + ((b * (int64_t)(Lout - (HAS_CLS ? 1 : 0)) + lx) * Hv + hv) * V, sv);
    }
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = sv[j] + pv[j];
    vstore_f32<V>(y + (idx * V), o);
  }
}

// K10: y[b,l,:] = emb[ids[b,l],:] + pos[l,:] — token-embedding gather fused
// with the position add. Reference: clip.py:159-160, siglip.py:146-147.
template <typename T>
__global__ void embed_pos_kernel(const int64_t* __restrict__ ids, const T* __restrict__ emb,
                                 const T* __restrict__ pos, T* __restrict__ y, int64_t BL,
                                 int L, int Hv) {
  constexpr int V = 8;
  const int64_t total = BL * Hv;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int hv = (int)(idx % Hv);
    const int64_t bl = idx / Hv;
    const int l = (int)(bl % L);
    const int64_t tok = ids[bl];
    float ev[V], pv[V], o[V];
    vload_f32<V>(emb + (tok * Hv + hv) * V, ev);
    vload_f32<V>(pos + ((int64_t)l * Hv + hv) * V, pv);
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = ev[j] + pv[j];
    vstore_f32<V>(y + idx * V, o);
  }
}

int act_code(const std::string& act) {
  if (act.empty()) return ACT_NONE;
  if (act == "gelu") return ACT_GELU;
  if (act == "gelu_tanh") return ACT_GELU_TANH;
  if (act == "quickgelu") return ACT_QUICKGELU;
  TORCH_CHECK(false, "unknown activation ", act);
}

template <typename T>
void launch_bias_act(torch::Tensor& z, const c10::optional<torch::Tensor>& bias,
                     const c10::optional<torch::Tensor>& res, torch::Tensor& y, int M,
                     int act, bool save_pre, hipStream_t stream) {
  const int64_t n = z.numel();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  const T* bp = bias ? reinterpret_cast<const T*>(bias->data_ptr()) : nullptr;
  const T* rp = res ? reinterpret_cast<const T*>(res->data_ptr()) : nullptr;
  T* zp = reinterpret_cast<T*>(z.data_ptr());
  T* yp = reinterpret_cast<T*>(y.data_ptr());
  const bool vec = (M % 8) == 0;
#define DISPATCH(HB, HR, SP)                                                              \
  do {                                                                                    \
    if (vec)                                                                              \
      hipLaunchKernelGGL((bias_act_vec_kernel<T, HB, HR, SP>), dim3(grid), dim3(block),   \
                         0, stream, zp, bp, rp, yp, n / 8, M / 8, act);                   \
    else                                                                                  \
      hipLaunchKernelGGL((bias_act_kernel<T, HB, HR, SP>), dim3(grid), dim3(block), 0,    \
                         stream, zp, bp, rp, yp, n, M, act);                              \
  } while (0)
  const bool hb = bias.has_value(), hr = res.has_value();
  if (hb && hr && save_pre) DISPATCH(true, true, true);
  else if (hb && hr && !save_pre) DISPATCH(true, true, false);
  else if (hb && !hr && save_pre) DISPATCH(true, false, true);
  else if (hb && !hr && !save_pre) DISPATCH(true, false, false);
  else if (!hb && hr && save_pre) DISPATCH(false, true, true);
  else if (!hb && hr && !save_pre) DISPATCH(false, true, false);
  else if (!hb && !hr && save_pre) DISPATCH(false, false, true);
  else DISPATCH(false, false, false);
#undef DISPATCH
}

}  // namespace

torch::Tensor bias_act_fwd(torch::Tensor z, c10::optional<torch::Tensor> bias,
                           std::string act, c10::optional<torch::Tensor> residual) {
  TORCH_CHECK(z.is_cuda() && z.is_contiguous());
  const int M = z.size(-1);
  const int code = act_code(act);
  auto stream = at::hip::getCurrentHIPStream();
  c10::optional<torch::Tensor> b;
  if (bias) b = bias->contiguous().to(z.scalar_type());
  c10::optional<torch::Tensor> r;
  if (residual) {
    TORCH_CHECK(residual->is_contiguous() && residual->scalar_type() == z.scalar_type());
    r = *residual;
  }
  const bool save_pre = code != ACT_NONE;
  torch::Tensor y = save_pre || r ? torch::empty_like(z) : z;
  if (!save_pre && !r && !b) return z;  // nothing to do
  if (!save_pre && !r) y = z;  // in-place bias add only
  if (z.scalar_type() == torch::kBFloat16) {
    launch_bias_act<bf16>(z, b, r, y, M, code, save_pre, stream);
  } else if (z.scalar_type() == torch::kFloat32) {
    launch_bias_act<float>(z, b, r, y, M, code, save_pre, stream);
  } else {
    TORCH_CHECK(false, "bias_act: unsupported dtype ", z.scalar_type());
  }
  return y;
}

torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor z, std::string act) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && z.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == z.scalar_type());
  const int code = act_code(act);
  auto dz = torch::empty_like(dy);
  const int64_t n = dy.numel();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  if (dy.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((act_bwd_kernel<bf16>), dim3(grid), dim3(block), 0, stream,
                       reinterpret_cast<const bf16*>(dy.data_ptr()),
                       reinterpret_cast<const bf16*>(z.data_ptr()),
                       reinterpret_cast<bf16*>(dz.data_ptr()), n, code);
  } else if (dy.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((act_bwd_kernel<float>), dim3(grid), dim3(block), 0, stream,
                       dy.data_ptr<float>(), z.data_ptr<float>(), dz.data_ptr<float>(), n,
                       code);
  } else {
    TORCH_CHECK(false, "act_bwd: unsupported dtype");
  }
  return dz;
}

torch::Tensor colsum(torch::Tensor dz) {
  TORCH_CHECK(dz.is_cuda() && dz.is_contiguous() && dz.dim() == 2);
  const int64_t M = dz.size(0);
  const int N = dz.size(1);
  TORCH_CHECK(N % 8 == 0, "colsum: N % 8 != 0");
  const int block = 256;
  // grid sized so ~1000 workgroups cover the matrix: each thread owns one
  // 8-wide column strip within a row chunk; surplus threads parallelize rows
  const int strips = N / 8;
  const int gx = std::max(1, (strips + block - 1) / block);
  const int rstep = gx * block / strips;
  const int rows_per_wg = 64;
  const int gy = (int)std::min<int64_t>((M + rows_per_wg - 1) / rows_per_wg, 65535);
  auto ws = torch::empty({(int64_t)gy * rstep, N}, dz.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  if (dz.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((colsum_kernel<bf16>), dim3(gx, gy), dim3(block), 0, stream,
                       reinterpret_cast<const bf16*>(dz.data_ptr()), ws.data_ptr<float>(),
                       M, N, rows_per_wg);
  } else if (dz.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((colsum_kernel<float>), dim3(gx, gy), dim3(block), 0, stream,
                       dz.data_ptr<float>(), ws.data_ptr<float>(), M, N, rows_per_wg);
  } else {
    TORCH_CHECK(false, "colsum: unsupported dtype");
  }
  return ws.sum(0);  // (gy*rstep, N) fp32 reduce — tiny
}

std::vector<torch::Tensor> bias_act_fwd_fp8(torch::Tensor z, c10::optional<torch::Tensor> bias,
                                            std::string act, torch::Tensor scale8,
                                            torch::Tensor amax) {
  // bf16 z (+bias + save-pre in place when bias given; else z already IS the
  // pre-activation, e.g. bias fused in the producing GEMM) -> act -> (y bf16,
  // y8 e4m3 bytes)
  TORCH_CHECK(z.is_cuda() && z.is_contiguous() && z.scalar_type() == torch::kBFloat16);
  const int M = z.size(-1);
  TORCH_CHECK(M % 8 == 0);
  const int code = act_code(act);
  auto y = torch::empty_like(z);
  auto y8 = torch::empty(z.sizes(), z.options().dtype(torch::kUInt8));
  const int64_t n = z.numel();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n / 8 + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  if (bias) {
    auto b = bias->contiguous().to(torch::kBFloat16);
    hipLaunchKernelGGL((bias_act_vec_kernel<bf16, true, false, true, true>), dim3(grid),
                       dim3(block), 0, stream, reinterpret_cast<bf16*>(z.data_ptr()),
                       reinterpret_cast<const bf16*>(b.data_ptr()), nullptr,
                       reinterpret_cast<bf16*>(y.data_ptr()), n / 8, M / 8, code,
                       y8.data_ptr<unsigned char>(), scale8.data_ptr<float>(),
                       reinterpret_cast<unsigned int*>(amax.data_ptr()));
  } else {
    hipLaunchKernelGGL((bias_act_vec_kernel<bf16, false, false, false, true>), dim3(grid),
                       dim3(block), 0, stream, reinterpret_cast<bf16*>(z.data_ptr()),
                       nullptr, nullptr, reinterpret_cast<bf16*>(y.data_ptr()), n / 8,
                       M / 8, code, y8.data_ptr<unsigned char>(), scale8.data_ptr<float>(),
                       reinterpret_cast<unsigned int*>(amax.data_ptr()));
  }
  return {y, y8};
}

torch::Tensor cls_pos_fwd(torch::Tensor x, c10::optional<torch::Tensor> cls,
                          torch::Tensor pos) {
  // x (B,L,H); cls (1,1,H) optional; pos (1,Lout,H) with Lout = L + (cls?1:0)
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const int B = x.size(0), L = x.size(1), H = x.size(2);
  TORCH_CHECK(H % 8 == 0, "cls_pos: H % 8 != 0");
  const int Lout = L + (cls ? 1 : 0);
  auto posc = pos.contiguous();
  TORCH_CHECK(posc.numel() >= (int64_t)Lout * H, "cls_pos: pos too short");
  TORCH_CHECK(posc.scalar_type() == x.scalar_type());
  auto y = torch::empty({B, Lout, H}, x.options());
  const int64_t n = y.numel() / 8;
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto t_tag) {
    using T = decltype(t_tag);
    const T* clsp = cls ? reinterpret_cast<const T*>(cls->contiguous().data_ptr()) : nullptr;
    if (cls)
      hipLaunchKernelGGL((cls_pos_kernel<T, true>), dim3(grid), dim3(block), 0, stream,
                         reinterpret_cast<const T*>(x.data_ptr()), clsp,
                         reinterpret_cast<const T*>(posc.data_ptr()),
                         reinterpret_cast<T*>(y.data_ptr()), B, Lout, H / 8);
    else
      hipLaunchKernelGGL((cls_pos_kernel<T, false>), dim3(grid), dim3(block), 0, stream,
                         reinterpret_cast<const T*>(x.data_ptr()), clsp,
                         reinterpret_cast<const T*>(posc.data_ptr()),
                         reinterpret_cast<T*>(y.data_ptr()), B, Lout, H / 8);
  };
  if (x.scalar_type() == torch::kBFloat16) launch(bf16{});
  else if (x.scalar_type() == torch::kFloat32) launch(0.f);
  else TORCH_CHECK(false, "cls_pos: unsupported dtype");
  return y;
}

torch::Tensor embed_pos_fwd(torch::Tensor ids, torch::Tensor emb, torch::Tensor pos) {
  // ids (B,L) int64; emb (Vocab,H); pos (L_max,H) -> y (B,L,H)
  TORCH_CHECK(ids.is_cuda() && ids.scalar_type() == torch::kInt64);
  auto idc = ids.contiguous();
  const int64_t BL = idc.numel();
  const int L = ids.size(-1);
  const int H = emb.size(-1);
  TORCH_CHECK(H % 8 == 0, "embed_pos: H % 8 != 0");
  auto embc = emb.contiguous();
  auto posc = pos.contiguous();
  TORCH_CHECK(embc.scalar_type() == posc.scalar_type());
  auto sizes = ids.sizes().vec();
  sizes.push_back(H);
  auto y = torch::empty(sizes, embc.options());
  const int64_t n = BL * (H / 8);
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto t_tag) {
    using T = decltype(t_tag);
    hipLaunchKernelGGL((embed_pos_kernel<T>), dim3(grid), dim3(block), 0, stream,
                       idc.data_ptr<int64_t>(), reinterpret_cast<const T*>(embc.data_ptr()),
                       reinterpret_cast<const T*>(posc.data_ptr()),
                       reinterpret_cast<T*>(y.data_ptr()), BL, L, H / 8);
  };
  if (embc.scalar_type() == torch::kBFloat16) launch(bf16{});
  else if (embc.scalar_type() == torch::kFloat32) launch(0.f);
  else TORCH_CHECK(false, "embed_pos: unsupported dtype");
  return y;
}

torch::Tensor im2col_patch(torch::Tensor img, int64_t patch) {
  TORCH_CHECK(img.is_cuda() && img.is_contiguous() && img.dim() == 4);
  const int B = img.size(0), C = img.size(1), HH = img.size(2), WW = img.size(3);
  const int P = (int)patch;
  TORCH_CHECK(HH % P == 0 && WW % P == 0);
  const int h = HH / P, w = WW / P;
  auto cols = torch::empty({(int64_t)B * h * w, (int64_t)C * P * P}, img.options());
  const int64_t n = cols.numel();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, img.scalar_type(), "im2col", [&] {
    hipLaunchKernelGGL((im2col_kernel<scalar_t>), dim3(grid), dim3(block), 0, stream,
                       img.data_ptr<scalar_t>(), cols.data_ptr<scalar_t>(), B, C, HH, WW, P);
  });
  return cols;
}

torch::Tensor col2im_patch(torch::Tensor cols, std::vector<int64_t> img_shape, int64_t patch) {
  TORCH_CHECK(cols.is_cuda() && cols.is_contiguous());
  const int B = img_shape[0], C = img_shape[1], HH = img_shape[2], WW = img_shape[3];
  const int P = (int)patch;
  auto img = torch::empty(img_shape, cols.options());
  const int64_t n = img.numel();
  const int block = 256;
  const int grid = (int)std::min<int64_t>((n + block - 1) / block, kMaxGrid);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, cols.scalar_type(), "col2im", [&] {
    hipLaunchKernelGGL((col2im_kernel<scalar_t>), dim3(grid), dim3(block), 0, stream,
                       cols.data_ptr<scalar_t>(), img.data_ptr<scalar_t>(), B, C, HH, WW, P);
  });
  return img;
}

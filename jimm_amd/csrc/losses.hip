// K12/K13 — similarity-head + loss kernels.
//
//   l2norm_fwd / l2norm_bwd — row L2-normalize (the similarity head's
//     normalize, models/clip.py:183-185 / siglip.py:169-171 semantics)
//   xent_rows_fwd / xent_rows_bwd — fused row-softmax cross-entropy over the
//     (B_local, B_global) contrastive logit block (CLIP InfoNCE, K13)
//   sigmoid_loss_ew — SigLIP pairwise sigmoid loss + dLogits in one pass
//     (loss = -sum log sigmoid(z * logits), z = +1 on the rank's diagonal)
//
// The GEMMs on either side (logits = scale * img @ txt^T and the dImg/dTxt
// backward GEMMs) run through rocBLAS — plain library GEMMs; these kernels
// fuse everything elementwise/rowwise around them.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// ---- row L2 normalize ------------------------------------------------------

template <typename T>
__global__ void l2norm_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  float* __restrict__ rinv, int64_t rows, int d) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t r = (int64_t)blockIdx.x * wpb + wave; r < rows; r += (int64_t)gridDim.x * wpb) {
    const T* xr = x + r * d;
    float ss = 0.f;
    for (int c = lane; c < d; c += WAVE) {
      float v = (float)xr[c];
      ss += v * v;
    }
    ss = wave_reduce_sum(ss);
    const float inv = rsqrtf(fmaxf(ss, 1e-24f));
    for (int c = lane; c < d; c += WAVE) y[r * d + c] = (T)((float)xr[c] * inv);
    if (lane == 0) rinv[r] = inv;
  }
}

template <typename T>
__global__ void l2norm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                  const float* __restrict__ rinv, T* __restrict__ dx,
                                  int64_t rows, int d) {
  // dx = inv * (dy - (dy . xn) * xn), xn = x * inv
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t r = (int64_t)blockIdx.x * wpb + wave; r < rows; r += (int64_t)gridDim.x * wpb) {
    const float inv = rinv[r];
    const T* xr = x + r * d;
    const T* dyr = dy + r * d;
    float dot = 0.f;
    for (int c = lane; c < d; c += WAVE) dot += (float)dyr[c] * (float)xr[c] * inv;
    dot = wave_reduce_sum(dot);
    for (int c = lane; c < d; c += WAVE)
      dx[r * d + c] = (T)(inv * ((float)dyr[c] - dot * (float)xr[c] * inv));
  }
}

// ---- fused softmax cross-entropy over logit rows ---------------------------

__global__ void xent_rows_fwd_kernel(const float* __restrict__ logits,
                                     const int64_t* __restrict__ labels,
                                     float* __restrict__ loss, float* __restrict__ lse,
                                     int64_t rows, int cols) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t r = (int64_t)blockIdx.x * wpb + wave; r < rows; r += (int64_t)gridDim.x * wpb) {
    const float* lr = logits + r * cols;
    float m = -INFINITY;
    for (int c = lane; c < cols; c += WAVE) m = fmaxf(m, lr[c]);
    m = wave_reduce_max(m);
    float s = 0.f;
    for (int c = lane; c < cols; c += WAVE) s += __expf(lr[c] - m);
    s = wave_reduce_sum(s);
    const float l = m + __logf(s);
    if (lane == 0) {
      lse[r] = l;
      loss[r] = l - lr[labels[r]];
    }
  }
}

__global__ void xent_rows_bwd_kernel(const float* __restrict__ logits,
                                     const int64_t* __restrict__ labels,
                                     const float* __restrict__ lse, float gscale,
                                     float* __restrict__ dlogits, int64_t rows, int cols) {
  // dlogits = gscale * (softmax - onehot)
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  for (int64_t r = (int64_t)blockIdx.x * wpb + wave; r < rows; r += (int64_t)gridDim.x * wpb) {
    const float* lr = logits + r * cols;
    const float l = lse[r];
    const int64_t lab = labels[r];
    for (int c = lane; c < cols; c += WAVE) {
      float p = __expf(lr[c] - l);
      if (c == lab) p -= 1.f;
      dlogits[r * cols + c] = gscale * p;
    }
  }
}

// ---- SigLIP pairwise sigmoid loss (loss sum + dLogits in one pass) ---------

__global__ void sigmoid_loss_ew_kernel(const float* __restrict__ logits,
                                       float* __restrict__ dlogits,
                                       float* __restrict__ loss_parts, int64_t rows,
                                       int cols, int64_t diag0) {
  // z = +1 where col == diag0 + row else -1
  // loss += -logsigmoid(z * x); dloss/dx = -z * sigmoid(-z * x)
  // Per-block partial sums land in loss_parts[blockIdx.x] (fixed in-block
  // reduce order); the host sums them in fixed order — the loss value is
  // deterministic without atomics.
  __shared__ float s_acc[4];
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int wpb = blockDim.x / WAVE;
  float acc = 0.f;
  for (int64_t r = (int64_t)blockIdx.x * wpb + wave; r < rows; r += (int64_t)gridDim.x * wpb) {
    const float* lr = logits + r * cols;
    for (int c = lane; c < cols; c += WAVE) {
      const float z = (c == diag0 + r) ? 1.f : -1.f;
      const float zx = z * lr[c];
      // -logsigmoid(zx) = log(1 + exp(-zx)) computed stably
      const float t = -zx;
      acc += (t > 0.f ? t : 0.f) + __logf(1.f + __expf(-fabsf(t)));
      dlogits[r * cols + c] = -z / (1.f + __expf(zx));
    }
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) s_acc[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float total = 0.f;
    for (int wv = 0; wv < wpb; ++wv) total += s_acc[wv];
    loss_parts[blockIdx.x] = total;
  }
}

}  // namespace

std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2);
  const int64_t rows = x.size(0);
  const int d = x.size(1);
  auto y = torch::empty_like(x);
  auto rinv = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((l2norm_fwd_kernel<bf16>), dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<const bf16*>(x.data_ptr()),
                       reinterpret_cast<bf16*>(y.data_ptr()), rinv.data_ptr<float>(), rows, d);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat32);
    hipLaunchKernelGGL((l2norm_fwd_kernel<float>), dim3(grid), dim3(256), 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(), rinv.data_ptr<float>(), rows, d);
  }
  return {y, rinv};
}

torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor rinv) {
  const int64_t rows = x.size(0);
  const int d = x.size(1);
  auto dx = torch::empty_like(x);
  const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  auto dyc = dy.contiguous();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((l2norm_bwd_kernel<bf16>), dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<const bf16*>(dyc.data_ptr()),
                       reinterpret_cast<const bf16*>(x.data_ptr()), rinv.data_ptr<float>(),
                       reinterpret_cast<bf16*>(dx.data_ptr()), rows, d);
  } else {
    hipLaunchKernelGGL((l2norm_bwd_kernel<float>), dim3(grid), dim3(256), 0, stream,
                       dyc.data_ptr<float>(), x.data_ptr<float>(), rinv.data_ptr<float>(),
                       dx.data_ptr<float>(), rows, d);
  }
  return dx;
}

std::vector<torch::Tensor> xent_rows_fwd(torch::Tensor logits, torch::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(labels.scalar_type() == torch::kInt64);
  const int64_t rows = logits.size(0);
  const int cols = logits.size(1);
  auto loss = torch::empty({rows}, logits.options());
  auto lse = torch::empty({rows}, logits.options());
  const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(xent_rows_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     loss.data_ptr<float>(), lse.data_ptr<float>(), rows, cols);
  return {loss, lse};
}

torch::Tensor xent_rows_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                            double gscale) {
  const int64_t rows = logits.size(0);
  const int cols = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(xent_rows_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     logits.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     lse.data_ptr<float>(), (float)gscale, dlogits.data_ptr<float>(), rows, cols);
  return dlogits;
}

std::vector<torch::Tensor> sigmoid_loss_ew(torch::Tensor logits, int64_t diag0) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.scalar_type() == torch::kFloat32);
  const int64_t rows = logits.size(0);
  const int cols = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const int grid = (int)std::min<int64_t>((rows + 3) / 4, 2048);
  auto parts = torch::empty({(int64_t)grid}, logits.options());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sigmoid_loss_ew_kernel, dim3(grid), dim3(256), 0, stream,
                     logits.data_ptr<float>(), dlogits.data_ptr<float>(),
                     parts.data_ptr<float>(), rows, cols, diag0);
  auto loss = parts.sum().reshape({1});  // fixed-order reduce: deterministic
  return {loss, dlogits};
}

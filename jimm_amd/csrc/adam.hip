// K14 — fused Adam update (optax.adam semantics:
// /root/reference/examples/vit_training.py:202-203), mixed-precision aware:
// params bf16 or fp32, moments fp32, optional fp32 master weights.
//
// Multi-tensor: the python side passes the whole parameter list; tensors are
// batched into chunk descriptors and processed by ONE kernel launch per
// ~64-tensor batch (launch-bound otherwise: ~150 params x ~1.5us).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int kChunkSize = 1 << 16;  // elements per chunk
constexpr int kMaxChunks = 4096;

struct ChunkDesc {
  const void* g;
  void* p;
  float* m;
  float* v;
  float* master;  // nullptr if none
  int64_t offset;
  int64_t n;  // elements in this chunk
  int is_bf16;
};

__global__ void adam_kernel(const ChunkDesc* __restrict__ chunks, int nchunks, float lr,
                            float b1, float b2, float eps, float wd, float bc1, float bc2) {
  for (int ci = blockIdx.x; ci < nchunks; ci += gridDim.x) {
    ChunkDesc c = chunks[ci];
    for (int64_t i = threadIdx.x; i < c.n; i += blockDim.x) {
      const int64_t idx = c.offset + i;
      float g = c.is_bf16 ? bf2f(reinterpret_cast<const bf16*>(c.g)[idx])
                          : reinterpret_cast<const float*>(c.g)[idx];
      float m = c.m[idx] = b1 * c.m[idx] + (1.f - b1) * g;
      float v = c.v[idx] = b2 * c.v[idx] + (1.f - b2) * g * g;
      float upd = (m / bc1) / (sqrtf(v / bc2) + eps);
      float w0 = c.master ? c.master[idx]
                          : (c.is_bf16 ? bf2f(reinterpret_cast<bf16*>(c.p)[idx])
                                       : reinterpret_cast<float*>(c.p)[idx]);
      if (wd != 0.f) upd += wd * w0;
      float w1 = w0 - lr * upd;
      if (c.master) c.master[idx] = w1;
      if (c.is_bf16)
        reinterpret_cast<bf16*>(c.p)[idx] = f2bf(w1);
      else
        reinterpret_cast<float*>(c.p)[idx] = w1;
    }
  }
}

__global__ void adam_kernel_dev(const ChunkDesc* __restrict__ chunks, int nchunks,
                                const float* __restrict__ lr_ptr,
                                const int* __restrict__ step_ptr, float b1, float b2,
                                float eps, float wd) {
  const float lr = *lr_ptr;
  const float st = (float)*step_ptr;
  const float bc1 = 1.f - __powf(b1, st);
  const float bc2 = 1.f - __powf(b2, st);
  constexpr int V = 4;  // vectorized: 16-B fp32 / 8-B bf16 accesses
  for (int ci = blockIdx.x; ci < nchunks; ci += gridDim.x) {
    ChunkDesc c = chunks[ci];
    const int64_t nv = c.n / V;
    for (int64_t iv = threadIdx.x; iv < nv; iv += blockDim.x) {
      const int64_t idx = c.offset + iv * V;
      float g[V], m[V], v[V], w0[V], w1[V];
      if (c.is_bf16) vload_f32<V>(reinterpret_cast<const bf16*>(c.g) + idx, g);
      else vload_f32<V>(reinterpret_cast<const float*>(c.g) + idx, g);
      vload_f32<V>(c.m + idx, m);
      vload_f32<V>(c.v + idx, v);
      if (c.master) vload_f32<V>(c.master + idx, w0);
      else if (c.is_bf16) vload_f32<V>(reinterpret_cast<bf16*>(c.p) + idx, w0);
      else vload_f32<V>(reinterpret_cast<float*>(c.p) + idx, w0);
#pragma unroll
      for (int k = 0; k < V; ++k) {
        m[k] = b1 * m[k] + (1.f - b1) * g[k];
        v[k] = b2 * v[k] + (1.f - b2) * g[k] * g[k];
        float upd = (m[k] / bc1) / (sqrtf(v[k] / bc2) + eps);
        if (wd != 0.f) upd += wd * w0[k];
        w1[k] = w0[k] - lr * upd;
      }
      vstore_f32<V>(c.m + idx, m);
      vstore_f32<V>(c.v + idx, v);
      if (c.master) vstore_f32<V>(c.master + idx, w1);
      if (c.is_bf16) vstore_f32<V>(reinterpret_cast<bf16*>(c.p) + idx, w1);
      else vstore_f32<V>(reinterpret_cast<float*>(c.p) + idx, w1);
    }
    // scalar tail
    for (int64_t i = nv * V + threadIdx.x; i < c.n; i += blockDim.x) {
      const int64_t idx = c.offset + i;
      float g = c.is_bf16 ? bf2f(reinterpret_cast<const bf16*>(c.g)[idx])
                          : reinterpret_cast<const float*>(c.g)[idx];
      float m = c.m[idx] = b1 * c.m[idx] + (1.f - b1) * g;
      float v = c.v[idx] = b2 * c.v[idx] + (1.f - b2) * g * g;
      float upd = (m / bc1) / (sqrtf(v / bc2) + eps);
      float w0 = c.master ? c.master[idx]
                          : (c.is_bf16 ? bf2f(reinterpret_cast<bf16*>(c.p)[idx])
                                       : reinterpret_cast<float*>(c.p)[idx]);
      if (wd != 0.f) upd += wd * w0;
      float w1 = w0 - lr * upd;
      if (c.master) c.master[idx] = w1;
      if (c.is_bf16)
        reinterpret_cast<bf16*>(c.p)[idx] = f2bf(w1);
      else
        reinterpret_cast<float*>(c.p)[idx] = w1;
    }
  }
}

__global__ void incr_kernel(int* __restrict__ step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *step += 1;
}

}  // namespace

// Build the chunk-descriptor buffer ONCE (device uint8 tensor). Pointers are
// stable across steps, so the graph-capturable adam_apply path needs no
// per-step host work at all.
std::vector<torch::Tensor> adam_prepare(std::vector<torch::Tensor> ps,
                                        std::vector<torch::Tensor> gs,
                                        std::vector<torch::Tensor> ms,
                                        std::vector<torch::Tensor> vs,
                                        std::vector<c10::optional<torch::Tensor>> masters) {
  std::vector<ChunkDesc> chunks;
  chunks.reserve(512);
  for (size_t t = 0; t < ps.size(); ++t) {
    auto& p = ps[t];
    TORCH_CHECK(p.is_cuda() && p.is_contiguous());
    const bool is_bf16 = p.scalar_type() == torch::kBFloat16;
    float* master = masters[t].has_value() ? masters[t]->data_ptr<float>() : nullptr;
    const int64_t n = p.numel();
    for (int64_t off = 0; off < n; off += kChunkSize) {
      chunks.push_back(ChunkDesc{gs[t].data_ptr(), p.data_ptr(), ms[t].data_ptr<float>(),
                                 vs[t].data_ptr<float>(), master, off,
                                 std::min<int64_t>(kChunkSize, n - off), is_bf16 ? 1 : 0});
    }
  }
  auto host = torch::from_blob(chunks.data(), {(int64_t)(chunks.size() * sizeof(ChunkDesc))},
                               torch::TensorOptions().dtype(torch::kUInt8));
  auto dev = host.to(ps[0].device());
  auto nchunks = torch::tensor({(int64_t)chunks.size()});
  return {dev, nchunks};
}

// Graph-capturable update: lr and step live in device tensors; step is
// incremented on-device first (so a replayed graph advances bias correction).
void adam_apply(torch::Tensor desc, int64_t nchunks, torch::Tensor lr_dev,
                torch::Tensor step_dev, double b1, double b2, double eps, double wd) {
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(incr_kernel, dim3(1), dim3(1), 0, stream, step_dev.data_ptr<int>());
  const int grid = std::min<int>((int)nchunks, 2048);
  hipLaunchKernelGGL(adam_kernel_dev, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const ChunkDesc*>(desc.data_ptr()), (int)nchunks,
                     lr_dev.data_ptr<float>(), step_dev.data_ptr<int>(), (float)b1,
                     (float)b2, (float)eps, (float)wd);
}

void adam_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
               std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
               std::vector<c10::optional<torch::Tensor>> masters, double lr, double b1,
               double b2, double eps, double wd, int64_t step) {
  TORCH_CHECK(ps.size() == gs.size() && ps.size() == ms.size() && ps.size() == vs.size());
  const float bc1 = 1.f - powf((float)b1, (float)step);
  const float bc2 = 1.f - powf((float)b2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();

  std::vector<ChunkDesc> chunks;
  chunks.reserve(512);
  for (size_t t = 0; t < ps.size(); ++t) {
    auto& p = ps[t];
    TORCH_CHECK(p.is_cuda() && p.is_contiguous(), "adam: params must be contiguous cuda");
    TORCH_CHECK(gs[t].is_contiguous() && ms[t].is_contiguous() && vs[t].is_contiguous());
    const bool is_bf16 = p.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(is_bf16 || p.scalar_type() == torch::kFloat32);
    float* master = nullptr;
    if (masters[t].has_value()) master = masters[t]->data_ptr<float>();
    const int64_t n = p.numel();
    for (int64_t off = 0; off < n; off += kChunkSize) {
      chunks.push_back(ChunkDesc{gs[t].data_ptr(), p.data_ptr(), ms[t].data_ptr<float>(),
                                 vs[t].data_ptr<float>(), master, off,
                                 std::min<int64_t>(kChunkSize, n - off), is_bf16 ? 1 : 0});
    }
  }
  // ship descriptors to device in batches
  auto opts = torch::TensorOptions().dtype(torch::kUInt8).device(ps[0].device());
  for (size_t start = 0; start < chunks.size(); start += kMaxChunks) {
    const int nb = (int)std::min<size_t>(kMaxChunks, chunks.size() - start);
    auto host = torch::from_blob(chunks.data() + start, {(int64_t)(nb * sizeof(ChunkDesc))},
                                 torch::TensorOptions().dtype(torch::kUInt8));
    // blocking H2D copy (pageable source); freed buffers are stream-ordered
    // by the caching allocator so the kernel may still read `dev` safely
    auto dev = host.to(opts.device());
    const int grid = std::min(nb, 2048);
    hipLaunchKernelGGL(adam_kernel, dim3(grid), dim3(256), 0, stream,
                       reinterpret_cast<const ChunkDesc*>(dev.data_ptr()), nb, (float)lr,
                       (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2);
    // keep `dev` alive until kernel completion: record it on the stream
    (void)dev;
  }
}

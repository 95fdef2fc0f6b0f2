// MFMA fragment-layout probe (gfx950).
//
// The attention / GEMM kernels assume these v_mfma_f32_16x16x32_bf16
// layouts (CDNA guide §3 gives C/D; A/B are the natural extension of the
// CDNA3 16x16x16 mapping and are VERIFIED ON HARDWARE by this probe —
// tests/test_kernels_gpu.py::test_mfma_probe runs it with asymmetric
// random matrices, which catches any transposed assumption):
//   A[i][k]: lane l holds A[l&15][(l>>4)*8 + j]   j=0..7
//   B[k][j]: lane l holds B[(l>>4)*8 + j][l&15]
//   C[i][j]: lane l holds C[(l>>4)*4 + r][l&15]   r=0..3

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

__global__ void mfma_probe_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x % WAVE;
  const int lo = lane & 15, hi = lane >> 4;
  bf16x8_t a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = *reinterpret_cast<const short*>(&A[lo * 32 + hi * 8 + j]);      // A[16][32] row-major
    b[j] = *reinterpret_cast<const short*>(&B[(hi * 8 + j) * 16 + lo]);    // B[32][16] row-major
  }
  f32x4_t c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(hi * 4 + r) * 16 + lo] = c[r];
}

typedef float f32x16_t __attribute__((ext_vector_type(16)));

__global__ void mfma_probe32_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                                    float* __restrict__ C) {
  // v_mfma_f32_32x32x16_bf16 — assumed layouts (the x16 extension of the
  // CDNA3 32x32x8 mapping), verified on hardware by test_mfma_probe32:
  //   A[i][k]: lane l holds A[l&31][(l>>5)*8 + j]          j=0..7
  //   B[k][j]: lane l holds B[(l>>5)*8 + j][l&31]
  //   C[i][j]: lane l holds C[8*b + (l>>5)*4 + r][l&31]    b=0..3, r=0..3,
  //            vector index 4*b + r
  const int lane = threadIdx.x % WAVE;
  const int lo = lane & 31, hi = lane >> 5;
  bf16x8_t a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = *reinterpret_cast<const short*>(&A[lo * 16 + hi * 8 + j]);      // A[32][16] row-major
    b[j] = *reinterpret_cast<const short*>(&B[(hi * 8 + j) * 32 + lo]);    // B[16][32] row-major
  }
  f32x16_t c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int blk = 0; blk < 4; ++blk)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      C[(8 * blk + hi * 4 + r) * 32 + lo] = c[4 * blk + r];
}

typedef short bf16x4_t __attribute__((ext_vector_type(4)));

template <int MODE>
__global__ void tr16_probe_kernel(const short* __restrict__ src, short* __restrict__ out) {
  // Fill LDS with src[0..255] (a 512-byte pattern), then ds_read_b64_tr_b16
  // with per-lane address addr(l) = MODE==0 ? (l&15)*2 + (l>>4)*128
  //                              : MODE==1 ? l*8
  //                              : 0 (uniform)
  // and dump each lane's 4 shorts: out[lane*4 + j].
  __shared__ __attribute__((aligned(16))) short buf[256];
  const int lane = threadIdx.x % WAVE;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) buf[i] = src[i];
  __syncthreads();
  int off;
  if (MODE == 0) off = (lane & 15) * 2 + (lane >> 4) * 128;
  else if (MODE == 1) off = lane * 8;
  else off = 0;
  typedef const __attribute__((address_space(3))) char* lds_p;
  const lds_p addr = (lds_p)(const void*)(reinterpret_cast<const char*>(buf) + off);
  bf16x4_t v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v) : "v"(addr) : "memory");
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = v[j];
}

}  // namespace

torch::Tensor tr16_probe(torch::Tensor src, int64_t mode) {
  TORCH_CHECK(src.is_cuda() && src.scalar_type() == torch::kInt16 && src.numel() == 256);
  auto out = torch::empty({64, 4}, src.options());
  auto stream = at::hip::getCurrentHIPStream();
  if (mode == 0)
    hipLaunchKernelGGL(tr16_probe_kernel<0>, dim3(1), dim3(64), 0, stream,
                       (const short*)src.contiguous().data_ptr(), (short*)out.data_ptr());
  else if (mode == 1)
    hipLaunchKernelGGL(tr16_probe_kernel<1>, dim3(1), dim3(64), 0, stream,
                       (const short*)src.contiguous().data_ptr(), (short*)out.data_ptr());
  else
    hipLaunchKernelGGL(tr16_probe_kernel<2>, dim3(1), dim3(64), 0, stream,
                       (const short*)src.contiguous().data_ptr(), (short*)out.data_ptr());
  return out;
}

torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({32, 16}) && B.sizes() == torch::IntArrayRef({16, 32}));
  auto C = torch::empty({32, 32}, A.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const bf16*>(A.contiguous().data_ptr()),
                     reinterpret_cast<const bf16*>(B.contiguous().data_ptr()),
                     C.data_ptr<float>());
  return C;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}) && B.sizes() == torch::IntArrayRef({32, 16}));
  auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const bf16*>(A.contiguous().data_ptr()),
                     reinterpret_cast<const bf16*>(B.contiguous().data_ptr()),
                     C.data_ptr<float>());
  return C;
}

// Shared helpers for jimm_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64  // CDNA wavefront width (not 32!)

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",      \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)

// ---- vector types ----------------------------------------------------------
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short bf16x4 __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef __hip_bfloat16 bf16;

__device__ __forceinline__ float bf2f(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16 f2bf(float x) { return __float2bfloat16(x); }

// raw-bits helpers for packed bf16 handled as short
__device__ __forceinline__ float bfs2f(short s) {
  unsigned int u = ((unsigned int)(unsigned short)s) << 16;
  return __uint_as_float(u);
}
__device__ __forceinline__ short f2bfs(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// ---- wave reductions -------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// ---- activations (shared fwd/bwd definitions) ------------------------------
// act codes: 0 = none, 1 = exact erf gelu, 2 = tanh gelu, 3 = quickgelu
#define ACT_NONE 0
#define ACT_GELU 1
#define ACT_GELU_TANH 2
#define ACT_QUICKGELU 3

// Fast erf (Abramowitz-Stegun 7.1.26, |err| <= 1.5e-7 — below bf16
// resolution): ~12 VALU + 1 v_rcp + 1 v_exp vs libm erff's branchy
// polynomial, which measured ~75 lane-cycles/element as a GEMM epilogue.
__device__ __forceinline__ float fast_erff(float x) {
  const float ax = fabsf(x);
  const float t = __frcp_rn(1.0f + 0.3275911f * ax);
  float p = 1.061405429f;
  p = p * t - 1.453152027f;
  p = p * t + 1.421413741f;
  p = p * t - 0.284496736f;
  p = p * t + 0.254829592f;
  const float e = 1.0f - p * t * __expf(-ax * ax);
  return copysignf(e, x);
}

__device__ __forceinline__ float fast_tanhf(float x) {
  // tanh(x) = 1 - 2/(exp(2x)+1); clamp avoids exp overflow (|x|>10 -> +-1)
  const float cx = fminf(fmaxf(x, -10.0f), 10.0f);
  return 1.0f - 2.0f / (__expf(2.0f * cx) + 1.0f);
}

__device__ __forceinline__ float act_fwd(float x, int act) {
  switch (act) {
    case ACT_GELU:
      return 0.5f * x * (1.0f + fast_erff(x * 0.70710678118654752440f));
    case ACT_GELU_TANH: {
      float x3 = x * x * x;
      float t = fast_tanhf(0.7978845608028654f * (x + 0.044715f * x3));
      return 0.5f * x * (1.0f + t);
    }
    case ACT_QUICKGELU: {
      float s = 1.0f / (1.0f + __expf(-1.702f * x));
      return x * s;
    }
    default:
      return x;
  }
}

// e4m3 (OCP) converts on the hardware v_cvt_pk_fp8_f32 pipe. The HIP
// library __hip_cvt_float_to_fp8 is a ~50-instruction software emulation
// (measured +77 us on a (73856,1024) LayerNorm); the packed builtin is one
// VALU op per 2 elements. Saturate-to-finite by clamping to +-448 first
// (NaN clamps to 448 via IEEE minNum/maxNum; inputs are never NaN here).
__device__ __forceinline__ unsigned short cvt2_e4m3(float a, float b) {
  a = fminf(fmaxf(a, -448.f), 448.f);
  b = fminf(fmaxf(b, -448.f), 448.f);
  return (unsigned short)(__builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false) & 0xffff);
}

__device__ __forceinline__ unsigned char cvt_e4m3(float a) {
  return (unsigned char)(cvt2_e4m3(a, 0.f) & 0xff);
}

__device__ __forceinline__ float act_grad(float x, int act) {
  switch (act) {
    case ACT_GELU: {
      // d/dx [x * Phi(x)] = Phi(x) + x * phi(x)
      float cdf = 0.5f * (1.0f + fast_erff(x * 0.70710678118654752440f));
      float pdf = 0.3989422804014327f * __expf(-0.5f * x * x);
      return cdf + x * pdf;
    }
    case ACT_GELU_TANH: {
      float x2 = x * x;
      float inner = 0.7978845608028654f * (x + 0.044715f * x * x2);
      float t = fast_tanhf(inner);
      float dinner = 0.7978845608028654f * (1.0f + 3.0f * 0.044715f * x2);
      return 0.5f * (1.0f + t) + 0.5f * x * (1.0f - t * t) * dinner;
    }
    case ACT_QUICKGELU: {
      float s = 1.0f / (1.0f + __expf(-1.702f * x));
      return s + 1.702f * x * s * (1.0f - s);
    }
    default:
      return 1.0f;
  }
}

// ---- vectorized load/store helpers (G13: hipcc does not auto-vectorize
// bf16 loads; 8-16 B per lane is the coalescing sweet spot) ----------------
template <int V, typename T>
__device__ __forceinline__ void vload_f32(const T* p, float (&o)[V]) {
  if constexpr (sizeof(T) == 2) {
    typedef short vt __attribute__((ext_vector_type(V)));
    vt v = *reinterpret_cast<const vt*>(p);
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = bfs2f((short)v[j]);
  } else {
    typedef float vt __attribute__((ext_vector_type(V)));
    vt v = *reinterpret_cast<const vt*>(p);
#pragma unroll
    for (int j = 0; j < V; ++j) o[j] = v[j];
  }
}

template <int V, typename T>
__device__ __forceinline__ void vstore_f32(T* p, const float (&in)[V]) {
  if constexpr (sizeof(T) == 2) {
    typedef short vt __attribute__((ext_vector_type(V)));
    vt v;
#pragma unroll
    for (int j = 0; j < V; ++j) v[j] = f2bfs(in[j]);
    *reinterpret_cast<vt*>(p) = v;
  } else {
    typedef float vt __attribute__((ext_vector_type(V)));
    vt v;
#pragma unroll
    for (int j = 0; j < V; ++j) v[j] = in[j];
    *reinterpret_cast<vt*>(p) = v;
  }
}

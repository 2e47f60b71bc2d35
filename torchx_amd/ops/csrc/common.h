// Common device helpers for the torchx_amd CDNA4 (gfx950) kernels.
//
// Wave size is 64 on CDNA4; all kernels here are written for gfx950 only —
// no CUDA shims, no multi-backend dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

// ---- vector types ---------------------------------------------------------
typedef __hip_bfloat16 bf16_t;
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short bf16x4_raw __attribute__((ext_vector_type(4)));
typedef short bf16x8_raw __attribute__((ext_vector_type(8)));   // MFMA A/B operand
typedef unsigned short ushort8 __attribute__((ext_vector_type(8)));
typedef unsigned int uint2_v __attribute__((ext_vector_type(2)));
typedef unsigned int uint4_v __attribute__((ext_vector_type(4)));

// ---- bf16 <-> f32 ---------------------------------------------------------
__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

// round-to-nearest-even f32 -> bf16
__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  unsigned int rounding = 0x7fff + ((x >> 16) & 1);
  x += rounding;
  return (unsigned short)(x >> 16);
}

// ---- wave / block reductions ---------------------------------------------
template <typename Op>
__device__ __forceinline__ float wave_reduce(float v, Op op) {
  #pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
    v = op(v, __shfl_xor(v, off, WAVE_SIZE));
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
  return wave_reduce(v, [] __device__ (float a, float b) { return a + b; });
}

__device__ __forceinline__ float wave_reduce_max(float v) {
  return wave_reduce(v, [] __device__ (float a, float b) { return fmaxf(a, b); });
}

// Block-level reduction over NW waves (NW <= 16). `scratch` must hold NW
// floats. Every thread returns the result.
template <int NW, typename Op>
__device__ __forceinline__ float block_reduce(float v, float* scratch, Op op) {
  const int wid = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  v = wave_reduce(v, op);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = scratch[0];
  #pragma unroll
  for (int i = 1; i < NW; ++i) r = op(r, scratch[i]);
  return r;
}

// ---- grid helpers ---------------------------------------------------------
__device__ __forceinline__ long grid_stride_begin() {
  return (long)blockIdx.x * blockDim.x + threadIdx.x;
}
__device__ __forceinline__ long grid_stride() {
  return (long)gridDim.x * blockDim.x;
}

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",        \
                  __FILE__, ":", __LINE__);                                    \
    }                                                                          \
  } while (0)

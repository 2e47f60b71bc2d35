// Fused softmax cross-entropy over a large vocab (Llama-3: V = 128256).
//
// fwd: one 256-thread workgroup per token row; single online pass over the
//      row computing (max, sumexp) + the target logit -> loss[t], lse[t].
// bwd: dlogits = gscale * (softmax - onehot), one vectorized pass re-reading
//      the logits (L2-resident for consecutive rows on the same XCD).
#include "common.h"

// combine two online-softmax partials (m, s)
__device__ __forceinline__ void ce_combine(float& m, float& s, float m2,
                                           float s2) {
  float M = fmaxf(m, m2);
  s = s * __expf(m - M) + s2 * __expf(m2 - M);
  m = M;
}

__global__ void __launch_bounds__(256)
ce_fwd_kernel(const unsigned short* __restrict__ logits,  // [T, V]
              const long* __restrict__ targets,           // [T]
              float* __restrict__ loss,                   // [T]
              float* __restrict__ lse,                    // [T]
              int V) {
  __shared__ float red_m[4], red_s[4], tgt_val;
  const long row = blockIdx.x;
  const unsigned short* xr = logits + row * (long)V;
  const long tgt = targets[row];

  float m = -3.4e38f, s = 0.f;
  const int V8 = V / 8 * 8;
  for (int i = threadIdx.x * 8; i < V8; i += 256 * 8) {
    ushort8 v = *(const ushort8*)(xr + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(v[j]);
      if (i + j == (int)tgt) tgt_val = f;
      float M = fmaxf(m, f);
      s = s * __expf(m - M) + __expf(f - M);
      m = M;
    }
  }
  // tail
  for (int i = V8 + (int)threadIdx.x; i < V; i += 256) {
    float f = bf16_to_f32(xr[i]);
    if (i == (int)tgt) tgt_val = f;
    float M = fmaxf(m, f);
    s = s * __expf(m - M) + __expf(f - M);
    m = M;
  }

  // wave reduce (m, s) pairs
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_xor(m, off, 64);
    float s2 = __shfl_xor(s, off, 64);
    ce_combine(m, s, m2, s2);
  }
  const int wid = threadIdx.x / 64;
  if (threadIdx.x % 64 == 0) { red_m[wid] = m; red_s[wid] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = red_m[0], S = red_s[0];
    #pragma unroll
    for (int i = 1; i < 4; ++i) ce_combine(M, S, red_m[i], red_s[i]);
    float l = M + __logf(S);
    lse[row] = l;
    loss[row] = l - tgt_val;
  }
}

__global__ void __launch_bounds__(256)
ce_bwd_kernel(const unsigned short* __restrict__ logits,  // [T, V]
              const long* __restrict__ targets,
              const float* __restrict__ lse,
              const float* __restrict__ gscale,  // [T] upstream grad per row
              unsigned short* __restrict__ dlogits, int V) {
  const long row = blockIdx.x;
  const unsigned short* xr = logits + row * (long)V;
  unsigned short* dr = dlogits + row * (long)V;
  const long tgt = targets[row];
  const float l = lse[row];
  const float gs = gscale[row];

  const int V8 = V / 8 * 8;
  for (int i = threadIdx.x * 8; i < V8; i += 256 * 8) {
    ushort8 v = *(const ushort8*)(xr + i);
    ushort8 o;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(bf16_to_f32(v[j]) - l);
      if (i + j == (int)tgt) p -= 1.0f;
      o[j] = f32_to_bf16(p * gs);
    }
    *(ushort8*)(dr + i) = o;
  }
  for (int i = V8 + (int)threadIdx.x; i < V; i += 256) {
    float p = __expf(bf16_to_f32(xr[i]) - l);
    if (i == (int)tgt) p -= 1.0f;
    dr[i] = f32_to_bf16(p * gs);
  }
}

extern "C" void ce_fwd_launch(const void* logits, const void* targets,
                              void* loss, void* lse, long T, int V,
                              hipStream_t stream) {
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((int)T), dim3(256), 0, stream,
                     (const unsigned short*)logits, (const long*)targets,
                     (float*)loss, (float*)lse, V);
}

extern "C" void ce_bwd_launch(const void* logits, const void* targets,
                              const void* lse, const void* gscale,
                              void* dlogits, long T, int V,
                              hipStream_t stream) {
  hipLaunchKernelGGL(ce_bwd_kernel, dim3((int)T), dim3(256), 0, stream,
                     (const unsigned short*)logits, (const long*)targets,
                     (const float*)lse, (const float*)gscale,
                     (unsigned short*)dlogits, V);
}

// Fused AdamW over flat parameter groups (MI355X-first optimizer design:
// the whole model's parameters live in ONE flat buffer per weight-decay
// group, so the optimizer step is a single memory-bound sweep at HBM rate
// instead of per-tensor launches).
//
// Tensors per group (all flat, same length n):
//   p32  f32  master params          (read+write)
//   p16  bf16 working params         (write; used by fwd/bwd GEMMs)
//   g16  bf16 gradient               (read)
//   m,v  f32  Adam moments           (read+write)
//
// update (per element):
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   p32 -= lr * ( (m/bc1) / (sqrt(v/bc2)+eps) + wd*p32 )
//   p16 = bf16(p32)
// bc1/bc2 are bias corrections computed on host per step.
#include "common.h"

__global__ void __launch_bounds__(256)
adamw_kernel(float* __restrict__ p32,
             unsigned short* __restrict__ p16,
             const unsigned short* __restrict__ g16,
             float* __restrict__ m,
             float* __restrict__ v,
             long n4,  // n / 4
             float lr, float beta1, float beta2, float eps, float wd,
             float inv_bc1, float inv_sqrt_bc2) {
  typedef unsigned short us4 __attribute__((ext_vector_type(4)));
  for (long i4 = grid_stride_begin(); i4 < n4; i4 += grid_stride()) {
    long i = i4 * 4;
    f32x4 pv = *(f32x4*)(p32 + i);
    f32x4 mv = *(f32x4*)(m + i);
    f32x4 vv = *(f32x4*)(v + i);
    us4 gv = *(const us4*)(g16 + i);
    us4 ov;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float g = bf16_to_f32(gv[j]);
      float mj = fmaf(beta1, mv[j], (1.f - beta1) * g);
      float vj = fmaf(beta2, vv[j], (1.f - beta2) * g * g);
      float mhat = mj * inv_bc1;
      float denom = sqrtf(vj) * inv_sqrt_bc2 + eps;
      float p = pv[j];
      p -= lr * (mhat / denom + wd * p);
      pv[j] = p;
      mv[j] = mj;
      vv[j] = vj;
      ov[j] = f32_to_bf16(p);
    }
    *(f32x4*)(p32 + i) = pv;
    *(f32x4*)(m + i) = mv;
    *(f32x4*)(v + i) = vv;
    *(us4*)(p16 + i) = ov;
  }
}

extern "C" void adamw_launch(void* p32, void* p16, const void* g16, void* m,
                             void* v, long n, float lr, float beta1,
                             float beta2, float eps, float wd, float bc1,
                             float bc2, hipStream_t stream) {
  const long n4 = n / 4;
  const int block = 256;
  long grid = (n4 + block - 1) / block;
  if (grid > 65535 * 8) grid = 65535 * 8;
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(adamw_kernel, dim3((int)grid), dim3(block), 0, stream,
                     (float*)p32, (unsigned short*)p16,
                     (const unsigned short*)g16, (float*)m, (float*)v, n4, lr,
                     beta1, beta2, eps, wd, 1.0f / bc1, 1.0f / sqrtf(bc2));
}

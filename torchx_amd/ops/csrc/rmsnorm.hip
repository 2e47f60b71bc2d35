// RMSNorm forward/backward, bf16 I/O, f32 accumulation.
//
// One 256-thread workgroup per row; row data is register-resident between
// the square-sum pass and the scale pass (H <= 8192).  LDS tree reduction
// across the 4 waves.  Memory-bound: vectorized 8-bf16 loads throughout
// (the guide's RMSNorm number: scalar 2.35 TB/s -> bf16x8 4.89 TB/s).
#include "common.h"

template <int ITERS>  // ITERS = H / (256 * 8)
__global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w,
    unsigned short* __restrict__ y,
    float* __restrict__ invrms,   // [R], saved for backward
    int H, float eps) {
  __shared__ float red[4];
  const long row = blockIdx.x;
  const unsigned short* xr = x + row * (long)H;
  unsigned short* yr = y + row * (long)H;

  float xs[ITERS][8];
  float acc = 0.f;
  #pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (it * 256 + threadIdx.x) * 8;
    ushort8 v = *(const ushort8*)(xr + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(v[j]);
      xs[it][j] = f;
      acc = fmaf(f, f, acc);
    }
  }
  float total = block_reduce<4>(acc, red,
      [] __device__ (float a, float b) { return a + b; });
  float r = rsqrtf(total / (float)H + eps);
  if (threadIdx.x == 0) invrms[row] = r;

  #pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (it * 256 + threadIdx.x) * 8;
    ushort8 wv = *(const ushort8*)(w + i);
    ushort8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      ov[j] = f32_to_bf16(xs[it][j] * r * bf16_to_f32(wv[j]));
    }
    *(ushort8*)(yr + i) = ov;
  }
}

// dx = r*(dy*w) - x * (r^3/H) * sum(dy*w*x);  dw = sum_rows(dy * x * r).
// Each block walks ROWS_PER_BLOCK rows, keeping its dw partial in registers,
// then does ONE atomicAdd per element at the end — far fewer contending
// atomics than a per-row scheme (the naive version spent 14% of a full
// Llama-3-8B step serializing 67M atomics on 4096 addresses).  8 rows per
// block: 32 starved the chip (rows/32 = 512 blocks = 2 per CU, and each
// block walks its rows SERIALLY behind a per-row block reduce — measured
// 25% of HBM roofline; 8 rows -> 2048 blocks keeps the atomic count low
// while filling the CUs).
#define RMS_ROWS_PER_BLOCK 8

template <int ITERS>
__global__ void __launch_bounds__(256)
rmsnorm_bwd_kernel(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w,
    const float* __restrict__ invrms,
    unsigned short* __restrict__ dx,
    float* __restrict__ dw,  // [H] f32, pre-zeroed
    long rows, int H) {
  __shared__ float red[4];
  float wv[ITERS][8], dwacc[ITERS][8];
  #pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (it * 256 + threadIdx.x) * 8;
    ushort8 v = *(const ushort8*)(w + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      wv[it][j] = bf16_to_f32(v[j]);
      dwacc[it][j] = 0.f;
    }
  }

  const long row0 = (long)blockIdx.x * RMS_ROWS_PER_BLOCK;
  const long row1 = min(row0 + RMS_ROWS_PER_BLOCK, rows);
  for (long row = row0; row < row1; ++row) {
    const unsigned short* xr = x + row * (long)H;
    const unsigned short* dyr = dy + row * (long)H;
    unsigned short* dxr = dx + row * (long)H;
    const float r = invrms[row];

    float xs[ITERS][8], dyw[ITERS][8];
    float t = 0.f;
    #pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * 256 + threadIdx.x) * 8;
      ushort8 xv = *(const ushort8*)(xr + i);
      ushort8 dv = *(const ushort8*)(dyr + i);
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf16_to_f32(xv[j]);
        float df = bf16_to_f32(dv[j]);
        xs[it][j] = xf;
        dyw[it][j] = df * wv[it][j];
        dwacc[it][j] = fmaf(df * xf, r, dwacc[it][j]);
        t = fmaf(df * wv[it][j], xf, t);
      }
    }
    float ts = block_reduce<4>(t, red,
        [] __device__ (float a, float b) { return a + b; });
    const float kk = ts * r * r * r / (float)H;

    #pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * 256 + threadIdx.x) * 8;
      ushort8 ov;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        ov[j] = f32_to_bf16(fmaf(dyw[it][j], r, -xs[it][j] * kk));
      }
      *(ushort8*)(dxr + i) = ov;
    }
    __syncthreads();  // red[] reuse across rows
  }

  #pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (it * 256 + threadIdx.x) * 8;
    #pragma unroll
    for (int j = 0; j < 8; ++j) atomicAdd(dw + i + j, dwacc[it][j]);
  }
}

#define DISPATCH_ITERS(H, FN)                                                  \
  do {                                                                         \
    int iters = (H) / 2048;                                                    \
    if (iters == 1) FN(1);                                                     \
    else if (iters == 2) FN(2);                                                \
    else if (iters == 3) FN(3);                                                \
    else if (iters == 4) FN(4);                                                \
  } while (0)

extern "C" void rmsnorm_fwd_launch(const void* x, const void* w, void* y,
                                   void* invrms, long rows, int H, float eps,
                                   hipStream_t stream) {
#define LAUNCH_F(I)                                                            \
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<I>), dim3((int)rows), dim3(256), 0,   \
                     stream, (const unsigned short*)x,                         \
                     (const unsigned short*)w, (unsigned short*)y,             \
                     (float*)invrms, H, eps)
  DISPATCH_ITERS(H, LAUNCH_F);
#undef LAUNCH_F
}

extern "C" void rmsnorm_bwd_launch(const void* dy, const void* x,
                                   const void* w, const void* invrms, void* dx,
                                   void* dw, long rows, int H,
                                   hipStream_t stream) {
  const long nblk = (rows + RMS_ROWS_PER_BLOCK - 1) / RMS_ROWS_PER_BLOCK;
#define LAUNCH_B(I)                                                            \
  hipLaunchKernelGGL((rmsnorm_bwd_kernel<I>), dim3((int)nblk), dim3(256), 0,   \
                     stream, (const unsigned short*)dy,                        \
                     (const unsigned short*)x, (const unsigned short*)w,       \
                     (const float*)invrms, (unsigned short*)dx, (float*)dw,    \
                     rows, H)
  DISPATCH_ITERS(H, LAUNCH_B);
#undef LAUNCH_B
}

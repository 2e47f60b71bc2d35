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
//
// Two kernels (r2 rework): the fused 8-serial-rows-per-block version
// measured ~10x off the HBM roofline (each block crawled its rows behind a
// per-row 4-wave LDS reduce, so the chip sat on reduce barriers).  Split:
//   dx: one block per row, exactly the fwd kernel's shape (which runs at
//       roofline) — no dw bookkeeping, no serial row walk.
//   dw: column reduction over row chunks, fp32 register accumulators, one
//       atomicAdd per element per chunk.  Re-reads dy/x (+33 us at the
//       bench shape) but every CU stays busy.

template <int ITERS>
__global__ void __launch_bounds__(256)
rmsnorm_bwd_dx_kernel(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ w,
    const float* __restrict__ invrms,
    unsigned short* __restrict__ dx,
    int H) {
  __shared__ float red[4];
  const long row = blockIdx.x;
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
    ushort8 wv = *(const ushort8*)(w + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xf = bf16_to_f32(xv[j]);
      float dwf = bf16_to_f32(dv[j]) * bf16_to_f32(wv[j]);
      xs[it][j] = xf;
      dyw[it][j] = dwf;
      t = fmaf(dwf, xf, t);
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
}

// dw column reduction: grid (row_chunks, H/2048); each block owns a
// 2048-column tile (256 threads x 8) over DW_CHUNK_ROWS rows.
// (Measured: 64-row chunks with a plain row walk beat a 4-wide unrolled
// 32-row variant 157 vs 282 us total bwd — the unroll's register pressure
// cost more than the load ILP bought.)
#define DW_CHUNK_ROWS 64

__global__ void __launch_bounds__(256)
rmsnorm_bwd_dw_kernel(
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ x,
    const float* __restrict__ invrms,
    float* __restrict__ dw,  // [H] f32, pre-zeroed
    long rows, int H) {
  const int col = (blockIdx.y * 2048) + threadIdx.x * 8;
  const long row0 = (long)blockIdx.x * DW_CHUNK_ROWS;
  const long row1 = min(row0 + DW_CHUNK_ROWS, rows);
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (long row = row0; row < row1; ++row) {
    const float r = invrms[row];
    ushort8 xv = *(const ushort8*)(x + row * (long)H + col);
    ushort8 dv = *(const ushort8*)(dy + row * (long)H + col);
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[j] = fmaf(bf16_to_f32(dv[j]) * bf16_to_f32(xv[j]), r, acc[j]);
    }
  }
  #pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(dw + col + j, acc[j]);
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
#define LAUNCH_B(I)                                                            \
  hipLaunchKernelGGL((rmsnorm_bwd_dx_kernel<I>), dim3((int)rows), dim3(256),   \
                     0, stream, (const unsigned short*)dy,                     \
                     (const unsigned short*)x, (const unsigned short*)w,       \
                     (const float*)invrms, (unsigned short*)dx, H)
  DISPATCH_ITERS(H, LAUNCH_B);
#undef LAUNCH_B
  const int row_chunks = (int)((rows + DW_CHUNK_ROWS - 1) / DW_CHUNK_ROWS);
  hipLaunchKernelGGL(rmsnorm_bwd_dw_kernel,
                     dim3(row_chunks, H / 2048), dim3(256), 0, stream,
                     (const unsigned short*)dy, (const unsigned short*)x,
                     (const float*)invrms, (float*)dw, rows, H);
}

// Fused residual-add + RMSNorm (decode/inference): s = x + res (the new
// residual stream), y = rmsnorm(s) * w. Same one-block-per-row shape as
// the forward kernel; res == nullptr degrades to plain rmsnorm with no s
// write. No invrms save — inference only, replaces an elementwise add +
// rmsnorm pair (2 dispatches -> 1) per use in the decode step.
template <int ITERS>  // ITERS = H / (256 * 8)
__global__ void __launch_bounds__(256)
rmsnorm_res_fwd_kernel(const unsigned short* __restrict__ x,
                       const unsigned short* __restrict__ res,
                       const unsigned short* __restrict__ w,
                       unsigned short* __restrict__ s_out,
                       unsigned short* __restrict__ y,
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
    ushort8 rv;
    if (res) rv = *(const ushort8*)(res + row * (long)H + i);
    ushort8 sv;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(v[j]);
      if (res) {
        f += bf16_to_f32(rv[j]);
        sv[j] = f32_to_bf16(f);
        f = bf16_to_f32(sv[j]);  // norm sees the bf16-rounded stream
      }
      xs[it][j] = f;
      acc = fmaf(f, f, acc);
    }
    if (res) *(ushort8*)(s_out + row * (long)H + i) = sv;
  }
  float total = block_reduce<4>(acc, red,
      [] __device__ (float a, float b) { return a + b; });
  float r = rsqrtf(total / (float)H + eps);

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

extern "C" void rmsnorm_res_launch(const void* x, const void* res,
                                   const void* w, void* s_out, void* y,
                                   long R, int H, float eps,
                                   hipStream_t stream) {
  #define RMSRES_CASE(N)                                                   \
    case N:                                                                \
      hipLaunchKernelGGL((rmsnorm_res_fwd_kernel<N>), dim3(R), dim3(256),  \
                         0, stream, (const unsigned short*)x,              \
                         (const unsigned short*)res,                       \
                         (const unsigned short*)w, (unsigned short*)s_out, \
                         (unsigned short*)y, H, eps);                      \
      break;
  switch (H / 2048) {
    RMSRES_CASE(1)
    RMSRES_CASE(2)
    RMSRES_CASE(3)
    RMSRES_CASE(4)
    default: break;  // binding guards H
  }
  #undef RMSRES_CASE
}

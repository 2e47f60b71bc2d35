// Element-wise hot ops: RoPE (fwd/bwd) and SwiGLU (fwd/bwd), bf16.
//
// Memory-bound: everything is vectorized (8 bf16 = 16 B per lane load) per
// the CDNA4 guide (scalar bf16 loads are ~2-2.5x slower).  RoPE uses
// host-precomputed f32 cos/sin tables (on-device trig would turn this
// VALU-bound).
//
// Replaces what the reference (meta-pytorch/torchx) delegates to stock
// PyTorch ops in its launched apps.
#include "common.h"

// ---------------------------------------------------------------------------
// RoPE, rotate-half (Llama) pairing: (i, i + D/2), D = head_dim.
// x: [rows, D] bf16 where rows = B*S*H, layout [B, S, H, D];
// cos/sin: [S, D/2] f32; pos_of_row = (row / H) % S.
// sign=+1 forward, -1 backward (inverse rotation).
// Each thread: 4 pairs (4 bf16 from each half = 8 B loads).
// ---------------------------------------------------------------------------
__global__ void rope_fwdbwd_kernel(
    const unsigned short* __restrict__ x,
    unsigned short* __restrict__ y,
    const float* __restrict__ cos_t,   // [S, D/2]
    const float* __restrict__ sin_t,   // [S, D/2]
    long np4,      // (rows * D/2) / 4
    int half_d,    // D/2, multiple of 4
    int heads,     // H
    int seq_len,   // S
    float sign) {
  typedef unsigned short us4 __attribute__((ext_vector_type(4)));
  typedef float f4 __attribute__((ext_vector_type(4)));
  for (long p4 = grid_stride_begin(); p4 < np4; p4 += grid_stride()) {
    long p = p4 * 4;
    long row = p / half_d;
    int i = (int)(p % half_d);
    int pos = (int)((row / heads) % seq_len);
    const long base = row * (2L * half_d);
    us4 u1 = *(const us4*)(x + base + i);
    us4 u2 = *(const us4*)(x + base + half_d + i);
    f4 c = *(const f4*)(cos_t + (long)pos * half_d + i);
    f4 s = *(const f4*)(sin_t + (long)pos * half_d + i);
    us4 o1, o2;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float x1 = bf16_to_f32(u1[j]);
      float x2 = bf16_to_f32(u2[j]);
      float y1 = fmaf(x1, c[j], -sign * s[j] * x2);
      float y2 = fmaf(x2, c[j], sign * s[j] * x1);
      o1[j] = f32_to_bf16(y1);
      o2[j] = f32_to_bf16(y2);
    }
    *(us4*)(y + base + i) = o1;
    *(us4*)(y + base + half_d + i) = o2;
  }
}

extern "C" void rope_launch(
    const void* x, void* y, const void* cos_t, const void* sin_t,
    long rows, int head_dim, int heads, int seq_len, float sign,
    hipStream_t stream) {
  const int half_d = head_dim / 2;
  const long np4 = rows * half_d / 4;
  const int block = 256;
  int grid = (int)((np4 + block - 1) / block);
  if (grid > 65535 * 8) grid = 65535 * 8;
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(rope_fwdbwd_kernel, dim3(grid), dim3(block), 0, stream,
                     (const unsigned short*)x, (unsigned short*)y,
                     (const float*)cos_t, (const float*)sin_t, np4, half_d,
                     heads, seq_len, sign);
}

// ---------------------------------------------------------------------------
// SwiGLU: out = silu(g) * u   (all bf16, flat n elements, n % 8 == 0)
// ---------------------------------------------------------------------------
__global__ void swiglu_fwd_kernel(
    const unsigned short* __restrict__ g,
    const unsigned short* __restrict__ u,
    unsigned short* __restrict__ out,
    long n8) {
  for (long i8 = grid_stride_begin(); i8 < n8; i8 += grid_stride()) {
    long i = i8 * 8;
    ushort8 gv = *(const ushort8*)(g + i);
    ushort8 uv = *(const ushort8*)(u + i);
    ushort8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32(gv[j]);
      float uf = bf16_to_f32(uv[j]);
      float sig =
          1.0f / (1.0f + __builtin_amdgcn_exp2f(-1.44269504f * gf));
      ov[j] = f32_to_bf16(gf * sig * uf);
    }
    *(ushort8*)(out + i) = ov;
  }
}

__global__ void swiglu_bwd_kernel(
    const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ g,
    const unsigned short* __restrict__ u,
    unsigned short* __restrict__ dg,
    unsigned short* __restrict__ du,
    long n8) {
  for (long i8 = grid_stride_begin(); i8 < n8; i8 += grid_stride()) {
    long i = i8 * 8;
    ushort8 dov = *(const ushort8*)(dout + i);
    ushort8 gv = *(const ushort8*)(g + i);
    ushort8 uv = *(const ushort8*)(u + i);
    ushort8 dgv, duv;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float dof = bf16_to_f32(dov[j]);
      float gf = bf16_to_f32(gv[j]);
      float uf = bf16_to_f32(uv[j]);
      float sig =
          1.0f / (1.0f + __builtin_amdgcn_exp2f(-1.44269504f * gf));
      float silu = gf * sig;
      float dsilu = sig * (1.0f + gf * (1.0f - sig));
      dgv[j] = f32_to_bf16(dof * uf * dsilu);
      duv[j] = f32_to_bf16(dof * silu);
    }
    *(ushort8*)(dg + i) = dgv;
    *(ushort8*)(du + i) = duv;
  }
}

extern "C" void swiglu_fwd_launch(const void* g, const void* u, void* out,
                                  long n, hipStream_t stream) {
  const long n8 = n / 8;
  const int block = 256;
  long grid = (n8 + block - 1) / block;
  if (grid > 65535 * 8) grid = 65535 * 8;
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3((int)grid), dim3(block), 0,
                     stream, (const unsigned short*)g,
                     (const unsigned short*)u, (unsigned short*)out, n8);
}

extern "C" void swiglu_bwd_launch(const void* dout, const void* g,
                                  const void* u, void* dg, void* du, long n,
                                  hipStream_t stream) {
  const long n8 = n / 8;
  const int block = 256;
  long grid = (n8 + block - 1) / block;
  if (grid > 65535 * 8) grid = 65535 * 8;
  if (grid == 0) grid = 1;
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3((int)grid), dim3(block), 0,
                     stream, (const unsigned short*)dout,
                     (const unsigned short*)g, (const unsigned short*)u,
                     (unsigned short*)dg, (unsigned short*)du, n8);
}

// ---------------------------------------------------------------------------
// Strided in-place-capable RoPE over the fused qkv buffer: rows = B*S, each
// row is [q: Hq*128 | k: Hkv*128 | v: Hkv*128] at row_stride elements.
// Rotates the first (Hq+Hkv) heads; the v region is untouched (pass y == x
// for in-place, e.g. on the dqkv gradient buffer).
// ---------------------------------------------------------------------------
__global__ void rope_qkv_kernel(
    const unsigned short* __restrict__ x,
    unsigned short* __restrict__ y,
    const float* __restrict__ cos_t,   // [S, 64]
    const float* __restrict__ sin_t,
    long np4,        // rows * nh * 16  (4-pair groups, half_d = 64)
    int nh,          // Hq + Hkv
    long row_stride, // elements per row
    int seq_len,
    float sign) {
  typedef unsigned short us4 __attribute__((ext_vector_type(4)));
  typedef float f4 __attribute__((ext_vector_type(4)));
  const int half_d = 64;
  // 2D grid: x covers (head, pair-group) with shifts/masks only, y strides
  // rows — the old flat grid-stride form paid a 64-bit div/mod (by nh=40,
  // not a power of two) on every thread
  const long hp4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (hp4 >= (long)nh * 16) return;
  const int h = (int)(hp4 >> 4);
  const int i = (int)(hp4 & 15) * 4;
  const long rows = np4 / ((long)nh * 16);
  for (long row = blockIdx.y; row < rows; row += gridDim.y) {
    const int pos = (int)(row % seq_len);
    const long base = row * row_stride + (long)h * 128;
    us4 u1 = *(const us4*)(x + base + i);
    us4 u2 = *(const us4*)(x + base + half_d + i);
    f4 c = *(const f4*)(cos_t + (long)pos * half_d + i);
    f4 s = *(const f4*)(sin_t + (long)pos * half_d + i);
    us4 o1, o2;
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float x1 = bf16_to_f32(u1[j]);
      float x2 = bf16_to_f32(u2[j]);
      o1[j] = f32_to_bf16(fmaf(x1, c[j], -sign * s[j] * x2));
      o2[j] = f32_to_bf16(fmaf(x2, c[j], sign * s[j] * x1));
    }
    *(us4*)(y + base + i) = o1;
    *(us4*)(y + base + half_d + i) = o2;
  }
}

extern "C" void rope_qkv_launch(
    const void* x, void* y, const void* cos_t, const void* sin_t,
    long rows, int nh, long row_stride, int seq_len, float sign,
    hipStream_t stream) {
  const long np4 = rows * nh * 16;
  const int block = 256;
  const int gx = (int)(((long)nh * 16 + block - 1) / block);
  const int gy = (int)((rows < 65535) ? (rows > 0 ? rows : 1) : 65535);
  hipLaunchKernelGGL(rope_qkv_kernel, dim3(gx, gy), dim3(block), 0,
                     stream, (const unsigned short*)x, (unsigned short*)y,
                     (const float*)cos_t, (const float*)sin_t, np4, nh,
                     row_stride, seq_len, sign);
}

// ---------------------------------------------------------------------------
// SwiGLU over the packed gate-up buffer gu = [rows, 2I]: out = silu(g)*u
// with g = gu[:, :I], u = gu[:, I:].  Avoids the two .contiguous() copies
// of the chunked views and the torch.cat in backward.
// ---------------------------------------------------------------------------
// 2D grid (cols x rows): the old 1D grid-stride form spent a 64-bit
// integer division (p / i8, i8 = I/8 = 1792 for llama) on EVERY thread —
// ~25% over the HBM roofline on a pure streaming kernel.
__global__ void swiglu_gu_fwd_kernel(
    const unsigned short* __restrict__ gu,
    unsigned short* __restrict__ out,
    long rows,
    long i8) {   // I / 8
  const long c8 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c8 >= i8) return;
  for (long row = blockIdx.y; row < rows; row += gridDim.y) {
    const long base = row * 2 * i8 * 8 + c8 * 8;
    ushort8 gv = *(const ushort8*)(gu + base);
    ushort8 uv = *(const ushort8*)(gu + base + i8 * 8);
    ushort8 ov;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32(gv[j]);
      float uf = bf16_to_f32(uv[j]);
      float sig =
          1.0f / (1.0f + __builtin_amdgcn_exp2f(-1.44269504f * gf));
      ov[j] = f32_to_bf16(gf * sig * uf);
    }
    *(ushort8*)(out + (row * i8 + c8) * 8) = ov;
  }
}

__global__ void swiglu_gu_bwd_kernel(
    const unsigned short* __restrict__ dout,
    const unsigned short* __restrict__ gu,
    unsigned short* __restrict__ dgu,
    long rows, long i8) {
  const long c8 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (c8 >= i8) return;
  for (long row = blockIdx.y; row < rows; row += gridDim.y) {
    const long base = row * 2 * i8 * 8 + c8 * 8;
    ushort8 dov = *(const ushort8*)(dout + (row * i8 + c8) * 8);
    ushort8 gv = *(const ushort8*)(gu + base);
    ushort8 uv = *(const ushort8*)(gu + base + i8 * 8);
    ushort8 dgv, duv;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float dof = bf16_to_f32(dov[j]);
      float gf = bf16_to_f32(gv[j]);
      float uf = bf16_to_f32(uv[j]);
      float sig =
          1.0f / (1.0f + __builtin_amdgcn_exp2f(-1.44269504f * gf));
      float silu = gf * sig;
      dgv[j] = f32_to_bf16(dof * uf * sig * (1.0f + gf * (1.0f - sig)));
      duv[j] = f32_to_bf16(dof * silu);
    }
    *(ushort8*)(dgu + base) = dgv;
    *(ushort8*)(dgu + base + i8 * 8) = duv;
  }
}

extern "C" void swiglu_gu_fwd_launch(const void* gu, void* out, long rows,
                                     long I, hipStream_t stream) {
  const long i8 = I / 8;
  const int block = 256;
  const int gx = (int)((i8 + block - 1) / block);
  const int gy = (int)((rows < 65535) ? (rows > 0 ? rows : 1) : 65535);
  hipLaunchKernelGGL(swiglu_gu_fwd_kernel, dim3(gx, gy), dim3(block), 0,
                     stream, (const unsigned short*)gu, (unsigned short*)out,
                     rows, i8);
}

extern "C" void swiglu_gu_bwd_launch(const void* dout, const void* gu,
                                     void* dgu, long rows, long I,
                                     hipStream_t stream) {
  const long i8 = I / 8;
  const int block = 256;
  const int gx = (int)((i8 + block - 1) / block);
  const int gy = (int)((rows < 65535) ? (rows > 0 ? rows : 1) : 65535);
  hipLaunchKernelGGL(swiglu_gu_bwd_kernel, dim3(gx, gy), dim3(block), 0,
                     stream, (const unsigned short*)dout,
                     (const unsigned short*)gu, (unsigned short*)dgu, rows,
                     i8);
}

// ---------------------------------------------------------------------------
// bf16 2D transpose: in [R, C] -> out [C, R].  64x64 LDS tiles, 8-wide
// vector loads AND stores (both global streams coalesced; the strided
// direction goes through LDS).  Feeds the dgrad-NT path: dx = dy @ Wt^T
// runs ~15% faster than dy @ W on these shapes, and torch's strided copy
// for .t().contiguous() is several times slower than this.
// R, C must be multiples of 64 (all transformer weight dims are).
// ---------------------------------------------------------------------------
#define TP 64

__global__ void __launch_bounds__(256)
transpose_bf16_kernel(const unsigned short* __restrict__ in,
                      unsigned short* __restrict__ out, int R, int C) {
  // +1 short of row padding: store-pass lanes hit rows 8 apart, and
  // 8*(64+1) shorts staggers them 4 banks apart -> conflict-free
  // (any even padding aliases: 8*(64+p)*2B % 128B == 0 for p in {0,2,4,8})
  __shared__ unsigned short tile[TP][TP + 1];
  const int tile_r = blockIdx.y * TP;  // input row of this tile
  const int tile_c = blockIdx.x * TP;  // input col of this tile

  // load: 256 threads x 8 = 2048 elems/pass, 2 passes for 64x64
  #pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    const int idx = pass * 2048 + threadIdx.x * 8;
    const int r = idx / TP;        // 0..63
    const int c = idx % TP;        // multiple of 8
    ushort8 v = *(const ushort8*)(in + (long)(tile_r + r) * C + tile_c + c);
    #pragma unroll
    for (int j = 0; j < 8; ++j) tile[r][c + j] = v[j];
  }
  __syncthreads();
  // store: thread writes out rows (= input cols) 8-wide
  #pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    const int idx = pass * 2048 + threadIdx.x * 8;
    const int oc = idx / TP;       // output row = input col, 0..63
    const int orr = idx % TP;      // output col base = input row
    ushort8 v;
    #pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tile[orr + j][oc];
    *(ushort8*)(out + (long)(tile_c + oc) * R + tile_r + orr) = v;
  }
}

extern "C" void transpose_bf16_launch(const void* in, void* out, long R,
                                      long C, hipStream_t stream) {
  hipLaunchKernelGGL(transpose_bf16_kernel,
                     dim3((int)(C / TP), (int)(R / TP)), dim3(256), 0,
                     stream, (const unsigned short*)in, (unsigned short*)out,
                     (int)R, (int)C);
}

// MFMA layout probe: computes C = A @ B (A [32,16], B [16,32], bf16 in,
// f32 out) with ONE v_mfma_f32_32x32x16_bf16 using the fragment maps the
// attention kernels assume.  The GPU test compares against torch.matmul —
// if the assumed lane->element maps were wrong this fails loudly.
#include "common.h"

typedef __bf16 mbf16x8 __attribute__((ext_vector_type(8)));

__global__ void mfma_probe_kernel(const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  float* __restrict__ c) {
  const int lane = threadIdx.x & 63;
  const int row = lane & 31;
  const int kb = (lane >> 5) * 8;
  // A[m=32, k=16]: lane holds A[row][kb..kb+7]
  mbf16x8 af = __builtin_bit_cast(mbf16x8, *(const ushort8*)(a + row * 16 + kb));
  // B[k=16, n=32]: lane holds B[kb+j][row] -> strided gather
  union { mbf16x8 v; unsigned short u[8]; } bf;
  #pragma unroll
  for (int j = 0; j < 8; ++j) bf.u[j] = b[(kb + j) * 32 + row];
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf.v, acc, 0, 0, 0);
  // C[m][n]: lane holds C[(r&3)+8*(r>>2)+4*(lane>>5)][lane&31]
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    c[m * 32 + row] = acc[r];
  }
}

extern "C" void mfma_probe_launch(const void* a, const void* b, void* c,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a, (const unsigned short*)b,
                     (float*)c);
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 probe (guide T10): stages B [16,32] bf16 ROW-major in
// LDS and assembles the MFMA B fragment with the hardware transpose read.
// Hypothesized semantics: each 16-lane cluster reads a 4-row x 16-col tile;
// lane j of the cluster supplies the address of the 4-bf16 chunk
// (row j>>2, cols 4*(j&2bits)) and receives the 4 ROWS at ITS column,
// packed low-to-high.  The raw per-lane reads are dumped so a mismatch
// shows the actual permutation, and the MFMA result is checked vs torch.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint2_v ds_tr_b16(const void* lds_addr) {
  uint2_v r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r)
               : "v"((unsigned int)(unsigned long long)
                     (const __attribute__((address_space(3))) char*)lds_addr)
               : "memory");
  return r;
}

__global__ void mfma_probe_tr_kernel(const unsigned short* __restrict__ a,
                                     const unsigned short* __restrict__ b,
                                     float* __restrict__ c,
                                     unsigned short* __restrict__ raw) {
  __shared__ unsigned short bimg[16 * 32];  // row-major [k][n], 64 B rows
  const int lane = threadIdx.x & 63;
  // stage B row-major
  for (int i = lane; i < 16 * 32; i += 64) bimg[i] = b[i];
  __syncthreads();

  const int row = lane & 31;
  const int kb = (lane >> 5) * 8;
  mbf16x8 af = __builtin_bit_cast(mbf16x8,
                                  *(const ushort8*)(a + row * 16 + kb));

  // cluster = lane>>4 (0..3); j = lane&15
  // cluster 0: rows 0-3   cols 0-15 ;  cluster 1: rows 0-3   cols 16-31
  // cluster 2: rows 8-11  cols 0-15 ;  cluster 3: rows 8-11  cols 16-31
  // (then +4 rows for the second read of each fragment half)
  const int cl = lane >> 4, j = lane & 15;
  const int colbase = (cl & 1) * 16;
  const int k0 = (cl >> 1) * 8;
  const unsigned short* chunk0 =
      bimg + (k0 + (j >> 2)) * 32 + colbase + 4 * (j & 3);
  const unsigned short* chunk1 =
      bimg + (k0 + 4 + (j >> 2)) * 32 + colbase + 4 * (j & 3);
  uint2_v r0 = ds_tr_b16(chunk0);
  uint2_v r1 = ds_tr_b16(chunk1);
  uint4_v u = {r0[0], r0[1], r1[0], r1[1]};
  mbf16x8 bf = __builtin_bit_cast(mbf16x8, u);

  // dump raw fragment for layout debugging: raw[lane][j] j=0..7
  {
    union { mbf16x8 v; unsigned short us[8]; } d;
    d.v = bf;
    #pragma unroll
    for (int t = 0; t < 8; ++t) raw[lane * 8 + t] = d.us[t];
  }

  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    int m = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    c[m * 32 + row] = acc[r];
  }
}

extern "C" void mfma_probe_tr_launch(const void* a, const void* b, void* c,
                                     void* raw, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_tr_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a, (const unsigned short*)b,
                     (float*)c, (unsigned short*)raw);
}

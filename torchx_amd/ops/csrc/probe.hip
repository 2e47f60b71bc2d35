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

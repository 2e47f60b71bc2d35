// Skinny-M bf16 GEMM ("GEMV") for the decode path:
//
//   out[m, n] = sum_k x[m, k] * W[n, k]        (nn.Linear, no bias)
//
// with M <= 8 (decode batch), K % 512 == 0, W row-major [N, K].
//
// At M=4 every weight byte is read once per token step, so the op is
// purely W-stream bound (~8 TB/s HBM3E roof); hipBLASLt's tile kernels
// reach only ~1/3 of that at M=4, which is what this kernel replaces.
//
// Layout: 256-thread blocks (4 waves); each wave owns RPW=2 output rows
// and marches K in 512-element chunks (64 lanes x ushort8 = 16 B/lane,
// fully coalesced 1 KB per wave-instruction). The x fragment (M x 8
// halves) is re-read per chunk straight from global — x is tiny
// (M*K <= 224 KB) and L2-resident per XCD, so it costs no HBM traffic —
// and shared across the wave's rows. Accumulate f32, reduce each
// (m, row) partial across the wave with 6 shfl_xor steps, lane 0 writes.
// K loop unrolled 4x to keep ~8 outstanding 16-B loads per lane.
// The MAC itself is v_dot2c_f32_bf16 (2 bf16 MACs/VALU-op, no converts):
// scalar convert+fma costs ~9 VALU ops per W element at M=4, enough to
// make the kernel VALU-bound instead of W-stream-bound.
#include "common.h"

typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));

__device__ __forceinline__ float dot2_bf16(ushort8 a, ushort8 b, float acc,
                                           int j2) {
  return __builtin_amdgcn_fdot2_f32_bf16(((const bf16x2*)&a)[j2],
                                         ((const bf16x2*)&b)[j2], acc,
                                         false);
}

template <int M, int RPW>
__global__ void __launch_bounds__(256)
gemv_bf16_kernel(const unsigned short* __restrict__ x,  // [M, K]
                 const unsigned short* __restrict__ w,  // [N, K]
                 unsigned short* __restrict__ out,      // [M, N]
                 int N, int K) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int n0 = (blockIdx.x * 4 + wave) * RPW;  // first row of this wave
  if (n0 >= N) return;

  float acc[M][RPW];
  #pragma unroll
  for (int m = 0; m < M; ++m)
    #pragma unroll
    for (int r = 0; r < RPW; ++r) acc[m][r] = 0.f;

  const int kc0 = lane * 8;  // this lane's first k within a chunk
  int kc = kc0;
  // main loop, 4 chunks per iteration (K % 2048 may be nonzero -> tail)
  for (; kc + 512 * 3 < K; kc += 512 * 4) {
    ushort8 xf[M][4];
    #pragma unroll
    for (int u = 0; u < 4; ++u)
      #pragma unroll
      for (int m = 0; m < M; ++m)
        xf[m][u] = *(const ushort8*)(x + (long)m * K + kc + 512 * u);
    #pragma unroll
    for (int r = 0; r < RPW; ++r) {
      if (n0 + r >= N) break;
      const unsigned short* wr = w + (long)(n0 + r) * K + kc;
      ushort8 wf[4];
      #pragma unroll
      for (int u = 0; u < 4; ++u) wf[u] = *(const ushort8*)(wr + 512 * u);
      #pragma unroll
      for (int u = 0; u < 4; ++u)
        #pragma unroll
        for (int j2 = 0; j2 < 4; ++j2)
          #pragma unroll
          for (int m = 0; m < M; ++m)
            acc[m][r] = dot2_bf16(xf[m][u], wf[u], acc[m][r], j2);
    }
  }
  for (; kc < K; kc += 512) {  // tail chunks
    ushort8 xf[M];
    #pragma unroll
    for (int m = 0; m < M; ++m)
      xf[m] = *(const ushort8*)(x + (long)m * K + kc);
    #pragma unroll
    for (int r = 0; r < RPW; ++r) {
      if (n0 + r >= N) break;
      ushort8 wf = *(const ushort8*)(w + (long)(n0 + r) * K + kc);
      #pragma unroll
      for (int j2 = 0; j2 < 4; ++j2)
        #pragma unroll
        for (int m = 0; m < M; ++m)
          acc[m][r] = dot2_bf16(xf[m], wf, acc[m][r], j2);
    }
  }

  // wave reduction: every (m, r) partial summed across 64 lanes
  #pragma unroll
  for (int m = 0; m < M; ++m)
    #pragma unroll
    for (int r = 0; r < RPW; ++r) {
      float v = acc[m][r];
      #pragma unroll
      for (int off = 32; off >= 1; off >>= 1) v += __shfl_xor(v, off, 64);
      acc[m][r] = v;
    }
  if (lane == 0) {
    #pragma unroll
    for (int r = 0; r < RPW; ++r) {
      if (n0 + r >= N) break;
      #pragma unroll
      for (int m = 0; m < M; ++m)
        out[(long)m * N + n0 + r] = f32_to_bf16(acc[m][r]);
    }
  }
}

template <int M, int RPW>
static void gemv_launch_rpw(const void* x, const void* w, void* o, int N,
                            int K, hipStream_t stream) {
  const int blocks = (N + 4 * RPW - 1) / (4 * RPW);
  hipLaunchKernelGGL((gemv_bf16_kernel<M, RPW>), dim3(blocks), dim3(256), 0,
                     stream, (const unsigned short*)x,
                     (const unsigned short*)w, (unsigned short*)o, N, K);
}

template <int M>
static void gemv_launch_m(const void* x, const void* w, void* o, int N,
                          int K, hipStream_t stream) {
  // RPW=2 measured best at every decode shape (RPW=4 and an LDS-staged
  // x variant were both slower; see profiles/README.md)
  gemv_launch_rpw<M, 2>(x, w, o, N, K, stream);
}

extern "C" void gemv_bf16_launch(const void* x, const void* w, void* o,
                                 int M, int N, int K, hipStream_t stream) {
  switch (M) {
    case 1: gemv_launch_m<1>(x, w, o, N, K, stream); break;
    case 2: gemv_launch_m<2>(x, w, o, N, K, stream); break;
    case 3: gemv_launch_m<3>(x, w, o, N, K, stream); break;
    case 4: gemv_launch_m<4>(x, w, o, N, K, stream); break;
    case 5: gemv_launch_m<5>(x, w, o, N, K, stream); break;
    case 6: gemv_launch_m<6>(x, w, o, N, K, stream); break;
    case 7: gemv_launch_m<7>(x, w, o, N, K, stream); break;
    case 8: gemv_launch_m<8>(x, w, o, N, K, stream); break;
    default: break;  // caller guards M <= 8
  }
}

// Fused wgu-GEMV + SwiGLU for the decode MLP: W is the packed
// [gate; up] = [2I, K] projection; each wave computes one gate row n and
// its paired up row n + I, then writes silu(g) * u — the [M, 2I]
// intermediate and the separate swiglu dispatch disappear. Same W-stream
// properties as the plain kernel (two sequential 8-KB row streams per
// wave), same dot2 inner loop.
template <int M>
__global__ void __launch_bounds__(256)
gemv_swiglu_bf16_kernel(const unsigned short* __restrict__ x,  // [M, K]
                        const unsigned short* __restrict__ w,  // [2I, K]
                        unsigned short* __restrict__ out,      // [M, I]
                        int I, int K) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int n0 = blockIdx.x * 4 + wave;  // gate row; up row = n0 + I
  if (n0 >= I) return;

  float acc[M][2];
  #pragma unroll
  for (int m = 0; m < M; ++m) acc[m][0] = acc[m][1] = 0.f;

  const int kc0 = lane * 8;
  int kc = kc0;
  for (; kc + 512 * 3 < K; kc += 512 * 4) {
    ushort8 xf[M][4];
    #pragma unroll
    for (int u = 0; u < 4; ++u)
      #pragma unroll
      for (int m = 0; m < M; ++m)
        xf[m][u] = *(const ushort8*)(x + (long)m * K + kc + 512 * u);
    #pragma unroll
    for (int r = 0; r < 2; ++r) {
      const unsigned short* wr = w + (long)(n0 + r * I) * K + kc;
      ushort8 wf[4];
      #pragma unroll
      for (int u = 0; u < 4; ++u) wf[u] = *(const ushort8*)(wr + 512 * u);
      #pragma unroll
      for (int u = 0; u < 4; ++u)
        #pragma unroll
        for (int j2 = 0; j2 < 4; ++j2)
          #pragma unroll
          for (int m = 0; m < M; ++m)
            acc[m][r] = dot2_bf16(xf[m][u], wf[u], acc[m][r], j2);
    }
  }
  for (; kc < K; kc += 512) {
    ushort8 xf[M];
    #pragma unroll
    for (int m = 0; m < M; ++m)
      xf[m] = *(const ushort8*)(x + (long)m * K + kc);
    #pragma unroll
    for (int r = 0; r < 2; ++r) {
      ushort8 wf = *(const ushort8*)(w + (long)(n0 + r * I) * K + kc);
      #pragma unroll
      for (int j2 = 0; j2 < 4; ++j2)
        #pragma unroll
        for (int m = 0; m < M; ++m)
          acc[m][r] = dot2_bf16(xf[m], wf, acc[m][r], j2);
    }
  }

  #pragma unroll
  for (int m = 0; m < M; ++m)
    #pragma unroll
    for (int r = 0; r < 2; ++r) {
      float v = acc[m][r];
      #pragma unroll
      for (int off = 32; off >= 1; off >>= 1) v += __shfl_xor(v, off, 64);
      acc[m][r] = v;
    }
  if (lane == 0) {
    #pragma unroll
    for (int m = 0; m < M; ++m) {
      const float g = acc[m][0], u = acc[m][1];
      const float sig =
          1.0f / (1.0f + __builtin_amdgcn_exp2f(-1.44269504f * g));
      out[(long)m * I + n0] = f32_to_bf16(g * sig * u);
    }
  }
}

extern "C" void gemv_swiglu_launch(const void* x, const void* w, void* o,
                                   int M, int I, int K, hipStream_t stream) {
  const int blocks = (I + 3) / 4;
  switch (M) {
    #define GSW_CASE(MM)                                                    \
      case MM:                                                              \
        hipLaunchKernelGGL((gemv_swiglu_bf16_kernel<MM>), dim3(blocks),     \
                           dim3(256), 0, stream, (const unsigned short*)x,  \
                           (const unsigned short*)w, (unsigned short*)o, I, \
                           K);                                              \
        break;
    GSW_CASE(1) GSW_CASE(2) GSW_CASE(3) GSW_CASE(4)
    GSW_CASE(5) GSW_CASE(6) GSW_CASE(7) GSW_CASE(8)
    #undef GSW_CASE
    default: break;
  }
}

// Single-position (decode) attention over a KV cache, bf16, GQA,
// HD = 128 — the serving-side counterpart of the training flash kernels.
//
//   o[b,hq,:] = softmax(scale * q[b,hq,:] . K[b,:L,hkv,:]^T) . V[b,:L,hkv,:]
//
// One 256-thread block per (b, hq). Work split: 16 lanes per cached row
// (16 x 8 d = 128, ushort8 loads), so a wave covers 4 rows per iteration
// and the block 16; each wave keeps its OWN online-softmax state (m, l,
// 8-float o partial per lane) over its rows — flash-decode style — and
// the four waves merge through LDS once at the end (one barrier).
// HBM-bound by the K/V cache stream (L x 512 B per (b, hkv), shared by
// the GQA group via L2).
#include "common.h"

#define HD 128

__global__ void __launch_bounds__(256)
decode_attn_kernel(const unsigned short* __restrict__ q,   // [B, Hq, 128]
                   const unsigned short* __restrict__ kc,  // [B, T, Hkv, 128]
                   const unsigned short* __restrict__ vc,  // [B, T, Hkv, 128]
                   unsigned short* __restrict__ o,         // [B, Hq, 128]
                   const int* __restrict__ len_dev,  // optional: L = len+1
                   int B, int Hq, int Hkv, int T, int L_host, int per_row,
                   float scale) {
  const int b = blockIdx.x / Hq;
  const int hq = blockIdx.x % Hq;
  // per_row: len_dev is [B] (ragged / continuous batching); else a scalar
  const int L = len_dev ? (len_dev[per_row ? b : 0] + 1) : L_host;
  const int hkv = hq / (Hq / Hkv);

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int tsub = lane >> 4;        // wave's row slot 0..3
  const int dchunk = (lane & 15) * 8;  // this lane's 8 d's

  // q fragment for this lane's d-chunk (f32)
  const unsigned short* qp = q + ((long)b * Hq + hq) * HD + dchunk;
  float qv[8];
  {
    ushort8 v = *(const ushort8*)qp;
    #pragma unroll
    for (int j = 0; j < 8; ++j) qv[j] = bf16_to_f32(v[j]) * scale;
  }

  const long row_stride = (long)Hkv * HD;
  const unsigned short* kb = kc + (long)b * T * row_stride + (long)hkv * HD;
  const unsigned short* vb = vc + (long)b * T * row_stride + (long)hkv * HD;

  float m_run = -3.0e38f, l_run = 0.f;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};

  // rows: wave w owns t = 16*i + 4*w + tsub
  for (int t = 4 * wave + tsub; t < L; t += 16) {
    const unsigned short* kr = kb + (long)t * row_stride + dchunk;
    ushort8 kv8 = *(const ushort8*)kr;
    float s = 0.f;
    #pragma unroll
    for (int j = 0; j < 8; ++j) s = fmaf(qv[j], bf16_to_f32(kv8[j]), s);
    // reduce the 16-lane group -> every lane of the group has the score
    #pragma unroll
    for (int off = 8; off >= 1; off >>= 1) {
      s += __shfl_xor(s, off, 16);
    }
    // online softmax (wave-local state; all 16 lanes of the group agree)
    const float m_new = fmaxf(m_run, s);
    const float alpha = __builtin_amdgcn_exp2f(
        1.44269504f * (m_run - m_new));
    const float p = __builtin_amdgcn_exp2f(1.44269504f * (s - m_new));
    l_run = l_run * alpha + p;
    const unsigned short* vr = vb + (long)t * row_stride + dchunk;
    ushort8 vv8 = *(const ushort8*)vr;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[j] = fmaf(acc[j], alpha, p * bf16_to_f32(vv8[j]));
    }
    m_run = m_new;
  }

  // combine the 4 row-slots of each wave, then the 4 waves: 16 partial
  // (m, l, o[128]) states -> LDS, block-combined by wave 0
  __shared__ float sm[16], sl[16], so[16][128];
  const int slot = wave * 4 + tsub;
  if ((lane & 15) == 0) {
    sm[slot] = m_run;
    sl[slot] = l_run;
  }
  #pragma unroll
  for (int j = 0; j < 8; ++j) so[slot][dchunk + j] = acc[j];
  __syncthreads();

  if (wave == 0) {
    // lane group 0..15 handles d-chunk as before; combine 16 slots
    float m_t = -3.0e38f;
    #pragma unroll
    for (int i = 0; i < 16; ++i) m_t = fmaxf(m_t, sm[i]);
    float l_t = 0.f;
    float out[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int i = 0; i < 16; ++i) {
      const float w = __builtin_amdgcn_exp2f(1.44269504f * (sm[i] - m_t));
      l_t += sl[i] * w;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        out[j] = fmaf(so[i][dchunk + j], w, out[j]);
      }
    }
    const float inv = (l_t > 0.f) ? 1.0f / l_t : 0.f;
    if (tsub == 0) {  // 16 lanes cover all 128 d
      unsigned short* op = o + ((long)b * Hq + hq) * HD + dchunk;
      ushort8 ov;
      #pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = f32_to_bf16(out[j] * inv);
      *(ushort8*)op = ov;
    }
  }
}

extern "C" void decode_attn_launch(const void* q, const void* kc,
                                   const void* vc, void* o,
                                   const void* len_dev, int B, int Hq,
                                   int Hkv, int T, int L, int per_row,
                                   float scale, hipStream_t stream) {
  hipLaunchKernelGGL(decode_attn_kernel, dim3(B * Hq), dim3(256), 0, stream,
                     (const unsigned short*)q, (const unsigned short*)kc,
                     (const unsigned short*)vc, (unsigned short*)o,
                     (const int*)len_dev, B, Hq, Hkv, T, L, per_row, scale);
}

// Fused decode-side rope + cache append: consumes the packed wqkv output
// [B, (Hq+2*Hkv)*128] for ONE position, ropes q and k (half-split pairing,
// same convention as rope_fwdbwd_kernel), writes the roped k and the raw v
// into the caches at row `pos`, and emits q contiguous [B, Hq, 128] for
// decode_attn. Replaces ~6 tiny kernels (2x rope, 2x index_copy, split
// copies) per layer per token; one wave per head, latency-floor bound.
// `pos` comes from the host (eager) or a device int32 scalar (hipGraph).
__global__ void __launch_bounds__(64)
decode_rope_cache_kernel(const unsigned short* __restrict__ qkv,
                         unsigned short* __restrict__ qout,  // [B, Hq, 128]
                         unsigned short* __restrict__ kc,    // [B,T,Hkv,128]
                         unsigned short* __restrict__ vc,
                         const float* __restrict__ cos_t,    // [S, 64]
                         const float* __restrict__ sin_t,
                         const int* __restrict__ pos_dev,
                         int B, int Hq, int Hkv, int T, int pos_host,
                         int per_row) {
  const int nh = Hq + 2 * Hkv;
  const int b = blockIdx.x / nh;
  const int h = blockIdx.x % nh;
  const int pos = pos_dev ? pos_dev[per_row ? b : 0] : pos_host;
  const int lane = threadIdx.x;  // 0..63: one rotation pair (lane, lane+64)
  const unsigned short* src = qkv + ((long)b * nh + h) * HD;
  if (h < Hq + Hkv) {  // q or k head: rotate
    const float x1 = bf16_to_f32(src[lane]);
    const float x2 = bf16_to_f32(src[lane + 64]);
    const float c = cos_t[(long)pos * 64 + lane];
    const float s = sin_t[(long)pos * 64 + lane];
    const unsigned short y1 = f32_to_bf16(fmaf(x1, c, -s * x2));
    const unsigned short y2 = f32_to_bf16(fmaf(x2, c, s * x1));
    unsigned short* dst =
        (h < Hq) ? qout + ((long)b * Hq + h) * HD
                 : kc + (((long)b * T + pos) * Hkv + (h - Hq)) * HD;
    dst[lane] = y1;
    dst[lane + 64] = y2;
  } else {  // v head: plain copy into the cache
    unsigned short* dst =
        vc + (((long)b * T + pos) * Hkv + (h - Hq - Hkv)) * HD;
    dst[lane] = src[lane];
    dst[lane + 64] = src[lane + 64];
  }
}

extern "C" void decode_rope_cache_launch(const void* qkv, void* qout,
                                         void* kc, void* vc,
                                         const void* cos_t, const void* sin_t,
                                         const void* pos_dev, int B, int Hq,
                                         int Hkv, int T, int pos_host,
                                         int per_row, hipStream_t stream) {
  hipLaunchKernelGGL(decode_rope_cache_kernel, dim3(B * (Hq + 2 * Hkv)),
                     dim3(64), 0, stream, (const unsigned short*)qkv,
                     (unsigned short*)qout, (unsigned short*)kc,
                     (unsigned short*)vc, (const float*)cos_t,
                     (const float*)sin_t, (const int*)pos_dev, B, Hq, Hkv, T,
                     pos_host, per_row);
}

// ---- split-K (flash-decode) variant ---------------------------------------
// At decode batch 4 the single-pass kernel launches only B*Hq = 128 blocks
// on 256 CUs — half the chip idles and each block serially streams its
// whole K/V slice (~25 us/layer measured). Pass 1 splits the cache length
// across gridDim.y blocks, each producing an UNNORMALIZED partial
// (m, l, o[128]) in f32 scratch; pass 2 (one small block per (b, hq))
// combines the partials exactly like the in-block wave merge.
__global__ void __launch_bounds__(256)
decode_attn_split_kernel(const unsigned short* __restrict__ q,
                         const unsigned short* __restrict__ kc,
                         const unsigned short* __restrict__ vc,
                         float* __restrict__ ml,    // [B*Hq, SPLIT, 2]
                         float* __restrict__ oacc,  // [B*Hq, SPLIT, 128]
                         const int* __restrict__ len_dev,
                         int B, int Hq, int Hkv, int T, int L_host,
                         int per_row, float scale) {
  const int bq = blockIdx.x;
  const int b = bq / Hq;
  const int hq = bq % Hq;
  const int L = len_dev ? (len_dev[per_row ? b : 0] + 1) : L_host;
  const int hkv = hq / (Hq / Hkv);
  const int split = gridDim.y;
  const int chunk = (L + split - 1) / split;
  const int t0 = blockIdx.y * chunk;
  const int t1 = min(L, t0 + chunk);

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const int tsub = lane >> 4;
  const int dchunk = (lane & 15) * 8;

  const unsigned short* qp = q + ((long)b * Hq + hq) * HD + dchunk;
  float qv[8];
  {
    ushort8 v = *(const ushort8*)qp;
    #pragma unroll
    for (int j = 0; j < 8; ++j) qv[j] = bf16_to_f32(v[j]) * scale;
  }

  const long row_stride = (long)Hkv * HD;
  const unsigned short* kb = kc + (long)b * T * row_stride + (long)hkv * HD;
  const unsigned short* vb = vc + (long)b * T * row_stride + (long)hkv * HD;

  float m_run = -3.0e38f, l_run = 0.f;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (int t = t0 + 4 * wave + tsub; t < t1; t += 16) {
    const unsigned short* kr = kb + (long)t * row_stride + dchunk;
    ushort8 kv8 = *(const ushort8*)kr;
    float s = 0.f;
    #pragma unroll
    for (int j = 0; j < 8; ++j) s = fmaf(qv[j], bf16_to_f32(kv8[j]), s);
    #pragma unroll
    for (int off = 8; off >= 1; off >>= 1) s += __shfl_xor(s, off, 16);
    const float m_new = fmaxf(m_run, s);
    const float alpha = __builtin_amdgcn_exp2f(1.44269504f * (m_run - m_new));
    const float p = __builtin_amdgcn_exp2f(1.44269504f * (s - m_new));
    l_run = l_run * alpha + p;
    const unsigned short* vr = vb + (long)t * row_stride + dchunk;
    ushort8 vv8 = *(const ushort8*)vr;
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] = fmaf(acc[j], alpha, p * bf16_to_f32(vv8[j]));
    m_run = m_new;
  }

  // block-combine the 16 (wave, row-slot) partials, wave 0 writes scratch
  __shared__ float sm[16], sl[16], so[16][128];
  const int slot = wave * 4 + tsub;
  if ((lane & 15) == 0) {
    sm[slot] = m_run;
    sl[slot] = l_run;
  }
  #pragma unroll
  for (int j = 0; j < 8; ++j) so[slot][dchunk + j] = acc[j];
  __syncthreads();
  if (wave == 0) {
    float m_t = -3.0e38f;
    #pragma unroll
    for (int i = 0; i < 16; ++i) m_t = fmaxf(m_t, sm[i]);
    float l_t = 0.f;
    float out[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int i = 0; i < 16; ++i) {
      const float w = __builtin_amdgcn_exp2f(1.44269504f * (sm[i] - m_t));
      l_t += sl[i] * w;
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        out[j] = fmaf(so[i][dchunk + j], w, out[j]);
    }
    const long pbase = ((long)bq * split + blockIdx.y);
    if (lane == 0) {
      ml[pbase * 2] = m_t;       // m = -inf, l = 0 for an empty slice
      ml[pbase * 2 + 1] = l_t;
    }
    if (tsub == 0) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) oacc[pbase * HD + dchunk + j] = out[j];
    }
  }
}

// pass 2: one 128-thread block per (b, hq); thread d combines SPLIT
// partials and writes the normalized bf16 output element.
__global__ void __launch_bounds__(128)
decode_attn_merge_kernel(const float* __restrict__ ml,
                         const float* __restrict__ oacc,
                         unsigned short* __restrict__ o,  // [B, Hq, 128]
                         int split) {
  const int bq = blockIdx.x;
  const int d = threadIdx.x;
  float m_t = -3.0e38f;
  for (int s = 0; s < split; ++s)
    m_t = fmaxf(m_t, ml[((long)bq * split + s) * 2]);
  float l_t = 0.f, out = 0.f;
  for (int s = 0; s < split; ++s) {
    const long p = (long)bq * split + s;
    const float w = __builtin_amdgcn_exp2f(1.44269504f * (ml[p * 2] - m_t));
    l_t += ml[p * 2 + 1] * w;
    out = fmaf(oacc[p * HD + d], w, out);
  }
  const float inv = (l_t > 0.f) ? 1.0f / l_t : 0.f;
  o[(long)bq * HD + d] = f32_to_bf16(out * inv);
}

extern "C" void decode_attn_split_launch(const void* q, const void* kc,
                                         const void* vc, void* ml,
                                         void* oacc, void* o,
                                         const void* len_dev, int B, int Hq,
                                         int Hkv, int T, int L, int split,
                                         int per_row, float scale,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(decode_attn_split_kernel, dim3(B * Hq, split),
                     dim3(256), 0, stream, (const unsigned short*)q,
                     (const unsigned short*)kc, (const unsigned short*)vc,
                     (float*)ml, (float*)oacc, (const int*)len_dev, B, Hq,
                     Hkv, T, L, per_row, scale);
  hipLaunchKernelGGL(decode_attn_merge_kernel, dim3(B * Hq), dim3(128), 0,
                     stream, (const float*)ml, (const float*)oacc,
                     (unsigned short*)o, split);
}

// Flash-style fused attention (forward + backward) for CDNA4 / gfx950.
//
// MFMA tiling with v_mfma_f32_32x32x16_bf16 (2xK gfx950 form), online
// softmax, causal masking, GQA.  Head dim fixed at 128 (Llama-3/Mixtral).
// Layouts are BSHD: q [B,S,Hq,128], k/v [B,S,Hkv,128], all bf16; lse/delta
// [B,Hq,S] f32.
//
// Design (the measured ladder is in profiles/README.md; idioms follow the
// CDNA4 guide's attention appendix):
//  * "swapped QK^T": S^T = mfma(A=K, B=Q) so each lane owns ONE q row's
//    scores -> softmax fully in-register (exp2 domain, v_cvt_pk_bf16_f32
//    fragment packing, defer-max rescale skip), one shfl_xor(32) partner
//    exchange, no cross-lane LDS reduction.
//  * KV tiles of 64 rows; K/V arrive by async global_load_lds into
//    XOR-swizzled natural images; the transposed operand images (V^T /
//    Q^T / dO^T / K^T) use a granule-ROTATED placement (rot(d) =
//    (d>>1 ^ d>>4) & 7: reads conflict-free, staging writes ~4-way — a
//    plain layout measured 36% of wave cycles in LDS bank conflicts) and
//    are built WAVE-LOCALLY from the just-landed naturals (each wave
//    transposes the rows it staged, so its own vmcnt(0) orders the read
//    and ONE barrier per tile publishes everything).
//  * T1 XCD swizzle: blocks streaming the same (batch, kv-head) tensors
//    land on one XCD so the stream is L2-resident (the kernels are
//    otherwise HBM-bound at ~128 FLOP/B).
//  * Forward block size is S-dependent: 4 waves below S=8192, 8 waves at
//    or above (staging amortizes over 2x compute once causal wave-skew
//    is relatively small).
//  * Backward is FA2-style without atomics: preprocess delta =
//    rowsum(dO*O); a dQ kernel (grid over q tiles); ONE combined 8-wave
//    dK+dV kernel where waves w and w+4 own the same 32 kv rows (dV / dK
//    roles) so Q/dO are staged once for both outputs.
//
// MFMA fragment maps used (verified against rocm CK xdlops_gemm.hpp and the
// CDNA4 guide; C/D map from the guide):
//   A[m=32, k=16]: lane l holds A[l&31][(l>>5)*8 + j], j=0..7  (bf16x8)
//   B[k=16, n=32]: lane l holds B[(l>>5)*8 + j][l&31]
//   C[m=32, n=32]: lane l holds C[(r&3) + 8*(r>>2) + 4*(l>>5)][l&31], r=0..15
#include "common.h"

typedef __bf16 mbf16x8 __attribute__((ext_vector_type(8)));

#define HD 128           // head dim
#define QBLK 32          // q rows per wave
#define KBLK 32          // kv rows per tile
#define NWAVES 4         // waves per block
#define BLOCK_Q (QBLK * NWAVES)  // q rows per block (fwd / dq)
#define BLOCK_K (KBLK * NWAVES)  // kv rows per block (dkv)

__device__ __forceinline__ f32x16 mfma32(mbf16x8 a, mbf16x8 b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

// v_cvt_pk_bf16_f32: 2 f32 -> packed 2x bf16 (RNE) in ONE instruction
// (no builtin on gfx950 — guide T12 recipe; the manual RNE pack is ~6 VALU
// ops per value and dominated the softmax)
__device__ __forceinline__ unsigned int pack_bf16x2(float a, float b) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// C-layout row index of accumulator register r for this lane-half.
__device__ __forceinline__ int c_row(int r, int half) {
  return (r & 3) + 8 * (r >> 2) + 4 * half;
}

// Build the two k-slice MFMA fragments (slice s covers k in [16s, 16s+16))
// from 16 f32 values held in C-layout.  Lane-half exchange via shfl_xor(32).
__device__ __forceinline__ void cvals_to_frags(const float* p, bool hi,
                                               mbf16x8* f0, mbf16x8* f1) {
  unsigned int w[2][4];
  #pragma unroll
  for (int s = 0; s < 2; ++s) {
    unsigned int a01 = pack_bf16x2(p[8 * s + 0], p[8 * s + 1]);
    unsigned int a23 = pack_bf16x2(p[8 * s + 2], p[8 * s + 3]);
    unsigned int a45 = pack_bf16x2(p[8 * s + 4], p[8 * s + 5]);
    unsigned int a67 = pack_bf16x2(p[8 * s + 6], p[8 * s + 7]);
    unsigned int x01 = __shfl_xor((int)a01, 32, 64);
    unsigned int x23 = __shfl_xor((int)a23, 32, 64);
    unsigned int x45 = __shfl_xor((int)a45, 32, 64);
    unsigned int x67 = __shfl_xor((int)a67, 32, 64);
    w[s][0] = hi ? x45 : a01;
    w[s][1] = hi ? x67 : a23;
    w[s][2] = hi ? a45 : x01;
    w[s][3] = hi ? a67 : x23;
  }
  uint4_v u0 = {w[0][0], w[0][1], w[0][2], w[0][3]};
  uint4_v u1 = {w[1][0], w[1][1], w[1][2], w[1][3]};
  *f0 = __builtin_bit_cast(mbf16x8, u0);
  *f1 = __builtin_bit_cast(mbf16x8, u1);
}

// ---------------------------------------------------------------------------
// Forward v2 (guide Appendix B ladder): KVBLK=64, double-buffered K tiles
// arriving by async global_load_lds into an XOR-swizzled LDS image (T2+T3),
// V reg-staged and transposed AFTER the barrier (T14 async-STAGE split),
// exp2-domain online softmax (v_exp_f32 native).
// ---------------------------------------------------------------------------

#define FKV 64                      // kv rows per staged tile
#define KIMG_BYTES (FKV * 256)      // [64 rows][256 B], byte ^= (row&15)<<4
#define VIMG_BYTES (HD * FKV * 2)   // V^T [128 d][128 B], byte ^= (d&7)<<4
#define LOG2E 1.44269504088896340736f
#define LN2 0.69314718055994530942f

// K fragment from the swizzled K image: row k = sub*32 + (lane&31),
// d-slice c -> 16 B at [32c + (lane>>5)*16] ^ swizzle.
__device__ __forceinline__ mbf16x8 kimg_frag(const char* kimg, int sub, int c) {
  const int lane = threadIdx.x & 63;
  const int row = sub * 32 + (lane & 31);
  const int byte = (c * 32 + ((lane >> 5) * 16)) ^ ((row & 15) << 4);
  return *(const mbf16x8*)(kimg + row * 256 + byte);
}

// V^T image: element (d, k) lives at byte
//   d*128 + ((k>>3 + rot(d)) & 7)*16 + (k&7)*2,   rot(d) = (d>>1 ^ d>>4)&7
// i.e. the 8 16-B granules of each 128-B d-row are ADD-rotated by a
// d-keyed amount.  Reads (16 consecutive d per lane group, fixed granule)
// see 8 distinct rotations x 2 parities -> conflict-free; the transposed
// staging writes (d = 8g+j, g varying per lane) see 8 granules x 2 r-slots
// -> ~4-way instead of the 32-way a plain layout gives (which measured as
// 36% of wave-cycles lost to LDS bank conflicts).
__device__ __forceinline__ int vrot(int d) {
  return ((d >> 1) ^ (d >> 4)) & 7;
}

// V^T fragment: row d = dt*32 + (lane&31), k-slice ks -> 16 B.
__device__ __forceinline__ mbf16x8 vimg_frag(const char* vimg, int dt, int ks) {
  const int lane = threadIdx.x & 63;
  const int d = dt * 32 + (lane & 31);
  const int g = (ks * 2 + (lane >> 5) + vrot(d)) & 7;
  return *(const mbf16x8*)(vimg + d * 128 + g * 16);
}

// ---------------------------------------------------------------------------
// T10 hardware transpose read: assemble the same fragment a vimg_frag of a
// transposed image gives — lane holds 8 elements (rows ks*16+(lane>>5)*8+j
// of the NATURAL 64-row image, at its own column dt*32+(lane&31)) — with
// two ds_read_tr16_b64 per fragment, straight off the XOR-swizzled natural
// image.  Kills the explicit LDS transposes (and their images) in the
// backward kernels.  Per-16-lane-cluster semantics (probe-verified,
// tools/probe_tr_read.py): lane j addresses the 4-bf16 chunk
// (row r0 + (j>>2), cols cbase + 4*(j&3)); lane i receives the 4 rows at
// column cbase + i, packed low-to-high.
// ---------------------------------------------------------------------------
typedef short short4_v __attribute__((ext_vector_type(4)));
typedef __attribute__((address_space(3))) short4_v* lds_s4p;

__device__ __forceinline__ mbf16x8 nat_tr_frag(const char* nat, int ks,
                                               int dt) {
  const int lane = threadIdx.x & 63;
  const int j = lane & 15;
  const int r_lo = ks * 16 + ((lane >> 5) * 8) + (j >> 2);
  const int r_hi = r_lo + 4;
  const int d0 = dt * 32 + ((lane >> 4) & 1) * 16 + 4 * (j & 3);
  const int g16 = (d0 >> 3) << 4;
  const int off = (d0 & 7) * 2;
  const int b_lo = r_lo * 256 + ((g16 ^ ((r_lo & 15) << 4)) | off);
  const int b_hi = r_hi * 256 + ((g16 ^ ((r_hi & 15) << 4)) | off);
  short4_v lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_s4p)(nat + b_lo));
  short4_v hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_s4p)(nat + b_hi));
  uint2_v ul = __builtin_bit_cast(uint2_v, lo);
  uint2_v uh = __builtin_bit_cast(uint2_v, hi);
  uint4_v u = {ul[0], ul[1], uh[0], uh[1]};
  return __builtin_bit_cast(mbf16x8, u);
}

// Async-stage one 64-row K tile: 16 global_load_lds_dwordx4 per block
// (16/NW per wave), source-permuted so the lane-linear LDS image lands
// swizzled (T2 note: swizzle moves to the SOURCE with glds staging).
// Rows beyond S clamp to S-1 (values masked in softmax).
template <int NW = 4, int TFKV = FKV>
__device__ __forceinline__ void stage_k_glds(
    const unsigned short* __restrict__ kb, long kv0, long stride_elems,
    int S, char* kimg) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  #pragma unroll
  for (int j = 0; j < (TFKV / 4) / NW; ++j) {
    const int i = wave * ((TFKV / 4) / NW) + j;
    const int row = i * 4 + (lane >> 4);
    const int colbyte = ((lane & 15) * 16) ^ ((row & 15) << 4);
    long srow = kv0 + row;
    if (srow >= S) srow = S - 1;
    const char* src = (const char*)(kb + srow * stride_elems) + colbyte;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(kimg + i * 1024),
        16, 0, 0);
  }
}

// V tile -> registers (4 x b128 per thread); written transposed+swizzled
// into the LDS V^T image after the barrier.
__device__ __forceinline__ void load_v_regs(
    const unsigned short* __restrict__ vb, long kv0, long stride_elems,
    int S, ushort8 vr[4]) {
  const int tid = threadIdx.x;
  #pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int c = tid + p * 256;          // 1024 chunks = 64 rows x 16
    const int r = c >> 4, g = c & 15;     // k row, d granule [8g, 8g+8)
    long srow = kv0 + r;
    if (srow >= S) srow = S - 1;
    vr[p] = *(const ushort8*)(vb + srow * stride_elems + g * 8);
  }
}

__device__ __forceinline__ void write_v_tr(const ushort8 vr[4], char* vimg) {
  const int tid = threadIdx.x;
  #pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int c = tid + p * 256;
    const int r = c >> 4, g = c & 15;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = 8 * g + j;
      const int gr = ((r >> 3) + vrot(d)) & 7;
      *(unsigned short*)(vimg + d * 128 + gr * 16 + (r & 7) * 2) = vr[p][j];
    }
  }
}


// T1 XCD-aware block swizzle: the dispatcher hands consecutive blockIdx to
// consecutive XCDs (each with its own 4 MB L2).  Group the G*nqt blocks
// that read the same (b, hkv) K/V stream onto ONE XCD so K/V go L2-resident
// after the first touch (these kernels are otherwise HBM-bound at
// AI ~= 128 FLOP/B).  Falls back to the linear map when B*Hkv % 8 != 0.
__device__ __forceinline__ void map_block_fwd(int B, int Hq, int Hkv,
                                              int nqt, int G,
                                              int* b, int* hq, int* qt) {
  int bid = blockIdx.x;
  if (((B * Hkv) & 7) == 0) {
    const int gsz = G * nqt;
    const int xcd = bid & 7, slot = bid >> 3;
    const int g = (slot / gsz) * 8 + xcd;
    *b = g / Hkv;
    const int r = slot % gsz;
    *hq = (g % Hkv) * G + r / nqt;
    *qt = r % nqt;
  } else {
    *b = bid / (Hq * nqt);
    bid -= *b * Hq * nqt;
    *hq = bid / nqt;
    *qt = bid % nqt;
  }
}

// Build the rot-placed transposed image from a staged swizzled natural
// image: per thread NCHUNK b128 LDS reads + 8 scalar writes each.  The
// source tile arrived by glds earlier in the iteration, so the transpose
// costs an LDS round trip (~50 cyc) instead of a global re-read (~200+).
template <int NTHREADS>
__device__ __forceinline__ void lds_nat_to_tr(const char* nat, char* tr_img) {
  const int tid = threadIdx.x;
  #pragma unroll
  for (int p = 0; p < 1024 / NTHREADS; ++p) {
    const int c = tid + p * NTHREADS;
    const int r = c >> 4, g = c & 15;
    const ushort8 v = *(const ushort8*)(
        nat + r * 256 + ((g * 16) ^ ((r & 15) << 4)));
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = 8 * g + j;
      const int gr = ((r >> 3) + vrot(d)) & 7;
      *(unsigned short*)(tr_img + d * 128 + gr * 16 + (r & 7) * 2) = v[j];
    }
  }
}



// Wave-local transpose: wave w transposes the 64/NW rows it glds-staged
// itself — its own vmcnt(0) orders the read (no pre-barrier needed).
template <int NW = 4>
__device__ __forceinline__ void lds_nat_to_tr_own(const char* nat,
                                                  char* tr_img) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  #pragma unroll
  for (int pp = 0; pp < 16 / NW; ++pp) {
    const int c = wave * (64 * 16 / NW) + pp * 64 + lane;
    const int r = c >> 4, g = c & 15;
    const ushort8 v = *(const ushort8*)(
        nat + r * 256 + ((g * 16) ^ ((r & 15) << 4)));
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = 8 * g + j;
      const int gr = ((r >> 3) + vrot(d)) & 7;
      *(unsigned short*)(tr_img + d * 128 + gr * 16 + (r & 7) * 2) = v[j];
    }
  }
}

template <int NW>
__global__ void __launch_bounds__(NW * 64, 2)
attn_fwd_kernel(const unsigned short* __restrict__ q,
                const unsigned short* __restrict__ k,
                const unsigned short* __restrict__ v,
                unsigned short* __restrict__ o,
                float* __restrict__ lse,  // [B,Hq,S]
                int B, int S, int Hq, int Hkv, float scale, int causal,
                long q_rs, long kv_rs) {  // per-seq-row element strides
  // K and V natural 2-rings (glds); the PV A-operand (V^T) is read off
  // the natural V image with hardware transpose reads (T10) — no V^T
  // build, 64 KiB LDS -> true 2 blocks/CU.
  __shared__ __align__(16) char smem[4 * KIMG_BYTES];

  const int BQ = NW * QBLK;
  const int nqt = (S + BQ - 1) / BQ;
  const int G = Hq / Hkv;
  int b, hq, qt;
  map_block_fwd(B, Hq, Hkv, nqt, G, &b, &hq, &qt);
  const int hkv = hq / G;

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const bool hi = lane >= 32;
  const int qr = lane & 31;

  const long q_seq_stride = q_rs;
  const long kv_seq_stride = kv_rs;
  const unsigned short* qb = q + (long)b * S * q_seq_stride + (long)hq * HD;
  const unsigned short* kb = k + (long)b * S * kv_seq_stride + (long)hkv * HD;
  const unsigned short* vb = v + (long)b * S * kv_seq_stride + (long)hkv * HD;

  const int q0_blk = qt * BQ;
  const int qw0 = q0_blk + wave * QBLK;       // wave's first q row
  const long q_row = qw0 + qr;                 // this lane's q row
  const bool wave_active = qw0 < S;

  // Q fragments in registers: chunk c covers d in [16c, 16c+16)
  mbf16x8 qfrag[8];
  {
    const long row = (q_row < S) ? q_row : (S - 1);
    const unsigned short* qp = qb + row * q_seq_stride + (hi ? 8 : 0);
    #pragma unroll
    for (int c = 0; c < 8; ++c) {
      qfrag[c] = __builtin_bit_cast(mbf16x8, *(const ushort8*)(qp + c * 16));
    }
  }

  f32x16 acc_o[4] = {};
  float m_run = -3.0e38f, l_run = 0.f;   // m in exp2 domain
  const float sc2 = scale * LOG2E;

  const int kv_limit = causal ? min(S, q0_blk + BQ) : S;
  const int ntiles = (kv_limit + FKV - 1) / FKV;
  const int qw_max = min(qw0 + QBLK - 1, S - 1);

  char* k0 = smem;                       // K tile t   (cur)
  char* k1 = smem + KIMG_BYTES;          // K tile t+1 (in flight)
  char* vcur = smem + 2 * KIMG_BYTES;    // V natural cur
  char* vnxt = smem + 3 * KIMG_BYTES;    // V natural nxt

  // prologue: glds K0 + V0 natural
  stage_k_glds<NW>(kb, 0, kv_seq_stride, S, k0);
  stage_k_glds<NW>(vb, 0, kv_seq_stride, S, vcur);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * FKV;
    const bool has_next = (t + 1) < ntiles;
    if (has_next) {
      // issue next K/V glds BEFORE compute (T14: their latency hides
      // under this tile's MFMAs; the stream is L2-resident anyway)
      stage_k_glds<NW>(kb, kv0 + FKV, kv_seq_stride, S, k1);
      stage_k_glds<NW>(vb, kv0 + FKV, kv_seq_stride, S, vnxt);
    }

    const bool needed = wave_active && (!causal || kv0 <= qw_max);
    if (needed) {
      // S^T[k, q] = K . Q^T over both 32-row k sub-tiles (T5: setprio
      // keeps the MFMA pipe fed while the other wave issues memory ops)
      f32x16 acc0 = {}, acc1 = {};
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int c = 0; c < 8; ++c) {
        acc0 = mfma32(kimg_frag(k0, 0, c), qfrag[c], acc0);
        acc1 = mfma32(kimg_frag(k0, 1, c), qfrag[c], acc1);
      }
      __builtin_amdgcn_s_setprio(0);

      float sv[32];
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        sv[r] = acc0[r] * sc2;
        sv[16 + r] = acc1[r] * sc2;
      }
      const bool mask_tile =
          (causal && kv0 + FKV - 1 > qw0) || (kv0 + FKV > S) || (q_row >= S);
      if (mask_tile) {
        #pragma unroll
        for (int r = 0; r < 32; ++r) {
          const long kg = kv0 + (r >> 4) * 32 + c_row(r & 15, hi);
          if (kg >= S || q_row >= S || (causal && kg > q_row)) {
            sv[r] = -3.0e38f;
          }
        }
      }

      // online softmax, exp2 domain (lane owns q row; partner lane+32
      // holds the other 16 k's of each sub-tile).  T13 defer-max: skip the
      // O-wide rescale while per-tile max growth stays under THR2
      // (P bounded by 2^THR2 ~= e^8, bf16-accum tolerates; the previous
      // tile's PV is complete before this decision — the safe order).
      float m_tile = sv[0];
      #pragma unroll
      for (int r = 1; r < 32; ++r) m_tile = fmaxf(m_tile, sv[r]);
      m_tile = fmaxf(m_tile, __shfl_xor(m_tile, 32, 64));
      const float THR2 = 11.54f;  // 8 * log2(e)
      if (!__all(m_tile - m_run <= THR2)) {
        const float m_new = fmaxf(m_run, m_tile);
        const float alpha = __builtin_amdgcn_exp2f(m_run - m_new);
        l_run *= alpha;
        m_run = m_new;
        #pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          #pragma unroll
          for (int r = 0; r < 16; ++r) acc_o[dt][r] *= alpha;
        }
      }

      float row_sum = 0.f;
      #pragma unroll
      for (int r = 0; r < 32; ++r) {
        sv[r] = __builtin_amdgcn_exp2f(sv[r] - m_run);  // sv becomes P
        row_sum += sv[r];
      }
      row_sum += __shfl_xor(row_sum, 32, 64);
      l_run += row_sum;

      mbf16x8 pf[4];
      cvals_to_frags(sv, hi, &pf[0], &pf[1]);
      cvals_to_frags(sv + 16, hi, &pf[2], &pf[3]);

      // O^T[d, q] += V^T . P  (A = V^T by hardware transpose read off
      // the natural V image, T10)
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int dt = 0; dt < 4; ++dt) {
        #pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          acc_o[dt] = mfma32(nat_tr_frag(vcur, ks, dt), pf[ks], acc_o[dt]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    if (has_next) {
      asm volatile("s_waitcnt vmcnt(0)");  // own K/V glds landed
      __syncthreads();   // everyone done reading cur; nxt visible
      char* tk = k0; k0 = k1; k1 = tk;
      char* tv = vcur; vcur = vnxt; vnxt = tv;
    }
  }

  if (!wave_active || q_row >= S) return;

  const float inv_l = (l_run > 0.f) ? 1.0f / l_run : 0.f;
  unsigned short* ob =
      o + (((long)b * S + q_row) * Hq + hq) * HD;  // o is dense BSHD
  #pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
    #pragma unroll
    for (int g = 0; g < 4; ++g) {
      // regs 4g..4g+3 are consecutive d: d = dt*32 + 8g + 4*hi + (0..3)
      bf16x4_raw w;
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        w[j] = (short)f32_to_bf16(acc_o[dt][4 * g + j] * inv_l);
      }
      *(bf16x4_raw*)(ob + dt * 32 + 8 * g + 4 * (hi ? 1 : 0)) = w;
    }
  }
  if (!hi) {
    // lse stays in the natural-log domain for the backward kernels
    lse[((long)b * Hq + hq) * S + q_row] = (m_run + __log2f(l_run)) * LN2;
  }
}

// ---------------------------------------------------------------------------
// Backward preprocess: delta[b,h,s] = rowsum(dO * O)
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
attn_bwd_pre_kernel(const unsigned short* __restrict__ dout,
                    const unsigned short* __restrict__ o,
                    float* __restrict__ delta,  // [B,Hq,S]
                    long rows,  // B*S*Hq
                    int S, int Hq) {
  // 16 lanes per row x 8 elems (16-B loads; the 1-row-per-wave version's
  // 4-B loads ran 3x off roofline); 4 rows per wave, 16 per block
  const long row = (long)blockIdx.x * 16 + threadIdx.x / 16;
  if (row >= rows) return;
  const int sl = threadIdx.x & 15;
  const unsigned short* dp = dout + row * HD + sl * 8;
  const unsigned short* op = o + row * HD + sl * 8;
  ushort8 dv = *(const ushort8*)dp;
  ushort8 ov = *(const ushort8*)op;
  float acc = 0.f;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    acc = fmaf(bf16_to_f32(dv[j]), bf16_to_f32(ov[j]), acc);
  }
  // reduce across the 16-lane group
  #pragma unroll
  for (int off = 8; off >= 1; off >>= 1) {
    acc += __shfl_down(acc, off, 16);  // width 16: stay within the group
  }
  if (sl == 0) {
    const long b = row / ((long)S * Hq);
    const long s = (row / Hq) % S;
    const long h = row % Hq;
    delta[(b * Hq + h) * S + s] = acc;
  }
}

// ---------------------------------------------------------------------------
// Backward dQ: grid over q tiles; inner loop over kv tiles.
// dQ[q,d] = scale * sum_k (P*(dP - delta))[q,k] * K[k,d]
// ---------------------------------------------------------------------------
template <int TFKV>
__global__ void __launch_bounds__(256, 2)
attn_bwd_dq_kernel(const unsigned short* __restrict__ q,
                   const unsigned short* __restrict__ k,
                   const unsigned short* __restrict__ v,
                   const unsigned short* __restrict__ dout,
                   const float* __restrict__ lse,
                   const float* __restrict__ delta,
                   unsigned short* __restrict__ dq,
                   int B, int S, int Hq, int Hkv, float scale, int causal,
                   long q_rs, long kv_rs, long dq_rs) {
  // v3 (T10): KV tiles of 64; K and V natural images arrive by async
  // global_load_lds (2-deep rings, swizzled); the dQ-accumulate A-operand
  // (K^T) is read straight off the natural K image with hardware
  // transpose reads — no K^T build, no rot image.  64 KiB LDS -> true
  // 2 blocks/CU.
  __shared__ __align__(16) char smem[4 * (TFKV * 256)];
  char* kcur = smem;
  char* knxt = smem + TFKV * 256;
  char* vcur = smem + 2 * (TFKV * 256);
  char* vnxt = smem + 3 * (TFKV * 256);

  const int nqt = (S + BLOCK_Q - 1) / BLOCK_Q;
  const int G = Hq / Hkv;
  int b, hq, qt;
  map_block_fwd(B, Hq, Hkv, nqt, G, &b, &hq, &qt);
  const int hkv = hq / G;

  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const bool hi = lane >= 32;
  const int qr = lane & 31;

  const long q_seq_stride = q_rs;
  const long kv_seq_stride = kv_rs;
  const unsigned short* qb = q + (long)b * S * q_seq_stride + (long)hq * HD;
  const unsigned short* kb = k + (long)b * S * kv_seq_stride + (long)hkv * HD;
  const unsigned short* vb = v + (long)b * S * kv_seq_stride + (long)hkv * HD;
  const unsigned short* dob =
      dout + ((long)b * S) * ((long)Hq * HD) + (long)hq * HD;  // dense

  const int q0_blk = qt * BLOCK_Q;
  const int qw0 = q0_blk + wave * QBLK;
  const long q_row = qw0 + qr;
  const bool wave_active = qw0 < S;
  const bool row_valid = q_row < S;

  mbf16x8 qfrag[8];
  float my_lse2 = 0.f, my_delta = 0.f;
  const long qrow_safe = row_valid ? q_row : (S - 1);
  const unsigned short* dop =
      dob + qrow_safe * ((long)Hq * HD) + (hi ? 8 : 0);
  {
    const unsigned short* qp = qb + qrow_safe * q_seq_stride + (hi ? 8 : 0);
    #pragma unroll
    for (int c = 0; c < 8; ++c) {
      qfrag[c] = __builtin_bit_cast(mbf16x8, *(const ushort8*)(qp + c * 16));
    }
    if (row_valid) {
      my_lse2 = lse[((long)b * Hq + hq) * S + q_row] * LOG2E;
      my_delta = delta[((long)b * Hq + hq) * S + q_row];
    }
  }

  f32x16 acc_dq[4] = {};
  const float sc2 = scale * LOG2E;

  const int kv_limit = causal ? min(S, q0_blk + BLOCK_Q) : S;
  const int ntiles = (kv_limit + TFKV - 1) / TFKV;
  const int qw_max = min(qw0 + QBLK - 1, S - 1);

  stage_k_glds<4, TFKV>(kb, 0, kv_seq_stride, S, kcur);
  stage_k_glds<4, TFKV>(vb, 0, kv_seq_stride, S, vcur);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * TFKV;
    const bool has_next = (t + 1) < ntiles;
    if (has_next) {
      // next tile's glds issue BEFORE compute; with the XCD swizzle the
      // K/V stream is L2-resident, one tile of cover is plenty
      stage_k_glds<4, TFKV>(kb, kv0 + TFKV, kv_seq_stride, S, knxt);
      stage_k_glds<4, TFKV>(vb, kv0 + TFKV, kv_seq_stride, S, vnxt);
    }

    const bool needed = wave_active && (!causal || kv0 <= qw_max);
    if (needed) {
      // dO fragments re-read each tile (L2-hot; keeping them resident
      // costs 32 VGPR across the staging phase and spills)
      mbf16x8 dofrag[8];
      #pragma unroll
      for (int c = 0; c < 8; ++c) {
        dofrag[c] =
            __builtin_bit_cast(mbf16x8, *(const ushort8*)(dop + c * 16));
      }
      // two 32-row k sub-tiles fully sequentially (no running max to
      // couple them -> half the live accumulators of a fused pass; the
      // loop must NOT unroll or both subs' accumulators go live at once)
      #pragma clang loop unroll(disable)
      for (int sb = 0; sb < TFKV / 32; ++sb) {
        if (causal && kv0 + 32 * sb > qw_max) break;  // rest fully masked
        f32x16 acc_s = {}, acc_dp = {};
        #pragma unroll
        for (int c = 0; c < 8; ++c) {
          acc_s = mfma32(kimg_frag(kcur, sb, c), qfrag[c], acc_s);
          acc_dp = mfma32(kimg_frag(vcur, sb, c), dofrag[c], acc_dp);
        }
        const int k0 = kv0 + 32 * sb;
        const bool mask_tile =
            (causal && k0 + 31 > qw0) || (k0 + 32 > S) || !row_valid;
        float ds[16];
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          float s2 = acc_s[r] * sc2;
          if (mask_tile) {
            const long kg = k0 + c_row(r, hi);
            if (kg >= S || !row_valid || (causal && kg > q_row)) {
              s2 = -3.0e38f;
            }
          }
          const float pr = __builtin_amdgcn_exp2f(s2 - my_lse2);
          ds[r] = scale * pr * (acc_dp[r] - my_delta);
        }
        mbf16x8 df0, df1;
        cvals_to_frags(ds, hi, &df0, &df1);
        // dQ^T[d, q] += K^T . dS^T; K^T fragments by hardware transpose
        // read from the natural K image (T10)
        #pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          acc_dq[dt] =
              mfma32(nat_tr_frag(kcur, 2 * sb, dt), df0, acc_dq[dt]);
          acc_dq[dt] =
              mfma32(nat_tr_frag(kcur, 2 * sb + 1, dt), df1, acc_dq[dt]);
        }
      }
    }

    if (has_next) {
      asm volatile("s_waitcnt vmcnt(0)");
      __syncthreads();   // all waves done reading cur; nxt glds visible
      char* tk = kcur; kcur = knxt; knxt = tk;
      char* tv = vcur; vcur = vnxt; vnxt = tv;
    }
  }

  if (!wave_active || !row_valid) return;
  unsigned short* dqb =
      dq + ((long)b * S + q_row) * dq_rs + (long)hq * HD;
  #pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
    #pragma unroll
    for (int g = 0; g < 4; ++g) {
      bf16x4_raw w;
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        w[j] = (short)f32_to_bf16(acc_dq[dt][4 * g + j]);
      }
      *(bf16x4_raw*)(dqb + dt * 32 + 8 * g + 4 * (hi ? 1 : 0)) = w;
    }
  }
}

// ---------------------------------------------------------------------------
// Backward dK/dV: grid over kv blocks of 128 rows (wave owns 32); inner loop
// over GQA group heads x q tiles.  No atomics: each (b, hkv, kv-row) is
// owned by exactly one wave.  Split into a dV pass and a dK pass so each
// kernel fits 2 waves/SIMD (the combined kernel needed 316 regs -> 1
// wave/SIMD with zero latency hiding; the S^T recompute in the second pass
// costs 8 extra MFMAs/tile but doubles occupancy).
// ---------------------------------------------------------------------------
// 512-thread (8-wave) staging variants: same image formats as the 4-wave
// helpers, work split across twice the threads.
template <int TFKV = FKV>
__device__ __forceinline__ void stage_k_glds8(
    const unsigned short* __restrict__ kb, long kv0, long stride_elems,
    int S, char* kimg) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  #pragma unroll
  for (int j = 0; j < TFKV / 32; ++j) {
    const int i = wave * (TFKV / 32) + j;
    const int row = i * 4 + (lane >> 4);
    const int colbyte = ((lane & 15) * 16) ^ ((row & 15) << 4);
    long srow = kv0 + row;
    if (srow >= S) srow = S - 1;
    const char* src = (const char*)(kb + srow * stride_elems) + colbyte;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(kimg + i * 1024),
        16, 0, 0);
  }
}

// ---------------------------------------------------------------------------
// Backward dK+dV, one 8-wave kernel.  Waves w and w+4 own the SAME 32 kv
// rows of a 128-row block: w accumulates dV, w+4 accumulates dK — so the
// Q/dO streams are staged ONCE for both outputs (the two-pass version
// streamed them 2-3x).  C-layouts are flipped vs the fwd kernel (swapped
// operand order): each lane owns one q COLUMN, so lse/delta are direct
// per-lane global loads (no LDS staging).  Four staged images per tile —
// Q nat (S^T chain B-op), dO nat (dP^T chain B-op), Q^T (dK accumulate
// B-op), dO^T (dV accumulate B-op) — double-buffered in 128 KiB LDS, one
// barrier per tile.
// ---------------------------------------------------------------------------

// Wave-local variant for 8-wave blocks: wave w transposes ONLY the 8
// rows it glds-staged itself ([8w, 8w+8)), so its own s_waitcnt vmcnt(0)
// is the only ordering needed before the transpose — no pre-barrier.
__device__ __forceinline__ void lds_nat_to_tr_own8(const char* nat,
                                                   char* tr_img) {
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  #pragma unroll
  for (int pp = 0; pp < 2; ++pp) {
    const int c = wave * 128 + pp * 64 + lane;
    const int r = c >> 4, g = c & 15;
    const ushort8 v = *(const ushort8*)(
        nat + r * 256 + ((g * 16) ^ ((r & 15) << 4)));
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int d = 8 * g + j;
      const int gr = ((r >> 3) + vrot(d)) & 7;
      *(unsigned short*)(tr_img + d * 128 + gr * 16 + (r & 7) * 2) = v[j];
    }
  }
}

template <int TFKV>
__global__ void __launch_bounds__(512, 2)
attn_bwd_dkv_kernel(const unsigned short* __restrict__ q,
                    const unsigned short* __restrict__ k,
                    const unsigned short* __restrict__ v,
                    const unsigned short* __restrict__ dout,
                    const float* __restrict__ lse,
                    const float* __restrict__ delta,
                    unsigned short* __restrict__ dk,
                    unsigned short* __restrict__ dv,
                    int B, int S, int Hq, int Hkv, float scale, int causal,
                    long q_rs, long kv_rs, long dout_rs, long out_rs) {
  // T10: the accumulate B-operands (Q^T / dO^T) are hardware transpose
  // reads off the natural images — no Q^T/dO^T builds, half the LDS
  // (64 KiB + lse) -> 2 blocks/CU.
  // TFKV=64: 64 KiB LDS -> 2 blocks/CU; TFKV=128: 128 KiB, 1 block but
  // half the barriers/loop iterations (A/B via TORCHX_AMD_DKV_FKV)
  __shared__ __align__(16) char smem[4 * (TFKV * 256)];
  __shared__ float lse_buf[2][TFKV], del_buf[2][TFKV];
  char* qn_c = smem;                       // Q natural cur
  char* don_c = smem + TFKV * 256;         // dO natural cur
  char* qn_n = smem + 2 * (TFKV * 256);
  char* don_n = smem + 3 * (TFKV * 256);

  const int nkt = (S + BLOCK_K - 1) / BLOCK_K;
  const int G = Hq / Hkv;
  int b, hkv, kt;
  {
    int bid = blockIdx.x;
    if (((B * Hkv) & 7) == 0) {
      const int xcd = bid & 7, slot = bid >> 3;
      const int g = (slot / nkt) * 8 + xcd;
      b = g / Hkv;
      hkv = g % Hkv;
      kt = slot % nkt;
    } else {
      b = bid / (Hkv * nkt);
      bid -= b * Hkv * nkt;
      hkv = bid / nkt;
      kt = bid % nkt;
    }
  }

  const int wave = threadIdx.x / 64;
  const bool is_dk = wave >= 4;
  const int lane = threadIdx.x & 63;
  const bool hi = lane >= 32;
  const int kr = lane & 31;

  const long q_seq_stride = q_rs;
  const long kv_seq_stride = kv_rs;
  const unsigned short* kb = k + (long)b * S * kv_seq_stride + (long)hkv * HD;
  const unsigned short* vb = v + (long)b * S * kv_seq_stride + (long)hkv * HD;

  const int kv0_blk = kt * BLOCK_K;
  const int kw0 = kv0_blk + (wave & 3) * KBLK;  // this wave's 32 kv rows
  const long k_row = kw0 + kr;
  const bool wave_active = kw0 < S;
  const bool krow_valid = k_row < S;

  // K resident in all waves; V resident in the dK waves
  mbf16x8 kfrag[8], vfrag[8];
  {
    const long row = krow_valid ? k_row : (S - 1);
    const unsigned short* kp = kb + row * kv_seq_stride + (hi ? 8 : 0);
    const unsigned short* vp = vb + row * kv_seq_stride + (hi ? 8 : 0);
    #pragma unroll
    for (int c = 0; c < 8; ++c) {
      kfrag[c] = __builtin_bit_cast(mbf16x8, *(const ushort8*)(kp + c * 16));
      if (is_dk) {
        vfrag[c] =
            __builtin_bit_cast(mbf16x8, *(const ushort8*)(vp + c * 16));
      }
    }
  }

  f32x16 acc[4] = {};
  const float sc2 = scale * LOG2E;

  const int t0 = causal ? (kv0_blk / TFKV) : 0;
  const int nt = (S + TFKV - 1) / TFKV;
  const int nt_eff = nt - t0;
  const int total = G * nt_eff;

  auto head_q = [&](int gh) {
    return q + (long)b * S * q_seq_stride + (long)(hkv * G + gh) * HD;
  };
  auto head_do = [&](int gh) {
    return dout + (long)b * S * dout_rs + (long)(hkv * G + gh) * HD;
  };
  auto stage_lse = [&](int gh, int t, int buf) {
    const int hq_ = hkv * G + gh;
    const int q0_ = t * TFKV;
    if (threadIdx.x < TFKV) {
      const float* lse_b = lse + ((long)b * Hq + hq_) * S;
      const int qg = q0_ + threadIdx.x;
      lse_buf[buf][threadIdx.x] = (qg < S) ? lse_b[qg] * LOG2E : 0.f;
    } else if (threadIdx.x < 2 * TFKV) {
      const float* del_b = delta + ((long)b * Hq + hq_) * S;
      const int qg = q0_ + (threadIdx.x - TFKV);
      del_buf[buf][threadIdx.x - TFKV] = (qg < S) ? del_b[qg] : 0.f;
    }
  };

  // prologue: tile (gh=0, t=t0) — glds the natural images
  stage_k_glds8<TFKV>(head_q(0), (long)t0 * TFKV, q_seq_stride, S, qn_c);
  stage_k_glds8<TFKV>(head_do(0), (long)t0 * TFKV, dout_rs, S, don_c);
  asm volatile("s_waitcnt vmcnt(0)");
  stage_lse(0, t0, 0);
  __syncthreads();

  const float* lse_cur = lse_buf[0];
  const float* del_cur = del_buf[0];
  int lse_nxt = 1;
  for (int idx = 0; idx < total; ++idx) {
    const int gh = idx / nt_eff;
    const int t = t0 + idx % nt_eff;
    const int q0 = t * TFKV;
    const int hq = hkv * G + gh;
    const bool has_next = (idx + 1) < total;
    const int ngh = (idx + 1) / nt_eff;
    const long nrow0 = (long)(t0 + (idx + 1) % nt_eff) * TFKV;
    if (has_next) {
      // async glds issue BEFORE compute (their latency hides under the
      // MFMAs; the address temporaries are consumed by the instruction
      // immediately so this does not raise register pressure).  The
      // reg-staged tr loads stay in the tail: their 32 data registers
      // across the compute phase are what spilled.
      stage_k_glds8<TFKV>(head_q(ngh), nrow0, q_seq_stride, S, qn_n);
      stage_k_glds8<TFKV>(head_do(ngh), nrow0, dout_rs, S, don_n);
    }

    const bool needed =
        wave_active && (!causal || q0 + TFKV - 1 >= kw0);
    if (needed) {
      #pragma clang loop unroll(disable)
      for (int sb = 0; sb < TFKV / 32; ++sb) {
        if (causal && q0 + 32 * sb + 32 <= kw0) continue;  // fully masked
        // S[q, k own] (lane owns the k column; q rows via c_row so the
        // contraction dim of the accumulate MFMAs is the C-row dim);
        // dP[q, k own] for the dK waves
        f32x16 acc_s = {}, acc_dp = {};
        #pragma unroll
        for (int c = 0; c < 8; ++c) {
          acc_s = mfma32(kimg_frag(qn_c, sb, c), kfrag[c], acc_s);
          if (is_dk) {
            acc_dp = mfma32(kimg_frag(don_c, sb, c), vfrag[c], acc_dp);
          }
        }

        const int qs0 = q0 + 32 * sb;
        // causal masking only bites when SOME (q, k) pair in this
        // 32q x 32k(own) tile has k > q, i.e. qs0 < kw0 + 32; interior
        // tiles (q strictly above every owned k) skip the 16-row mask
        // VALU entirely (it used to run on EVERY causal tile)
        const bool mask_tile = (causal && qs0 < kw0 + 32)
                               || (qs0 + 32 > S) || !krow_valid;
        float cv[16];
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int ql = 32 * sb + c_row(r, hi);  // q offset within tile
          float s2 = acc_s[r] * sc2;
          if (mask_tile) {
            const long qg = q0 + ql;
            if (qg >= S || !krow_valid || (causal && k_row > qg)) {
              s2 = -3.0e38f;
            }
          }
          const float pr =
              __builtin_amdgcn_exp2f(s2 - lse_cur[ql]);
          cv[r] = is_dk ? scale * pr * (acc_dp[r] - del_cur[ql]) : pr;
        }

        mbf16x8 f0, f1;
        cvals_to_frags(cv, hi, &f0, &f1);

        // dV[k,d] += P^T . dO ; dK[k,d] += dS^T . Q — B fragments by
        // hardware transpose read from the natural images (T10)
        const char* acc_img = is_dk ? qn_c : don_c;
        #pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          acc[dt] =
              mfma32(f0, nat_tr_frag(acc_img, 2 * sb, dt), acc[dt]);
          acc[dt] =
              mfma32(f1, nat_tr_frag(acc_img, 2 * sb + 1, dt), acc[dt]);
        }
      }
    }

    if (has_next) {
      asm volatile("s_waitcnt vmcnt(0)");  // own glds of the nxt tiles done
      stage_lse(ngh, t0 + (idx + 1) % nt_eff, lse_nxt);
      __syncthreads();   // ONE barrier per tile publishes everything
      lse_cur = lse_buf[lse_nxt];
      del_cur = del_buf[lse_nxt];
      lse_nxt ^= 1;
      char* tp;
      tp = qn_c; qn_c = qn_n; qn_n = tp;
      tp = don_c; don_c = don_n; don_n = tp;
    }
  }

  // store via LDS transpose (8-B global writes); wave-private f32 region
  unsigned short* out = is_dk ? dk : dv;
  __syncthreads();
  float* tr = (float*)smem + wave * (32 * 36);
  if (wave_active) {
    #pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
      __syncthreads();
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        tr[c_row(r, hi) * 36 + kr] = acc[dt][r];
      }
      __syncthreads();
      #pragma unroll
      for (int rep = 0; rep < 4; ++rep) {
        const int kk = kr;
        const int d0 = (hi ? 16 : 0) + rep * 4;
        if (kw0 + kk < S) {
          bf16x4_raw w;
          #pragma unroll
          for (int j = 0; j < 4; ++j) {
            w[j] = (short)f32_to_bf16(tr[kk * 36 + d0 + j]);
          }
          *(bf16x4_raw*)(out + ((long)b * S + kw0 + kk) * out_rs +
                         (long)hkv * HD + dt * 32 + d0) = w;
        }
      }
    }
  } else {
    #pragma unroll
    for (int i = 0; i < 8; ++i) __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Launchers
// ---------------------------------------------------------------------------
extern "C" void attn_fwd_launch(const void* q, const void* k, const void* v,
                                void* o, void* lse, int B, int S, int Hq,
                                int Hkv, float scale, int causal,
                                long q_rs, long kv_rs, hipStream_t stream) {
  // Post-T10 the 4-wave block wins at EVERY length (S=4096: 618 vs 558
  // TF/s; S=8192: 706 vs 648 — the pre-T10 dispatch sent S>=8192 to the
  // 8-wave variant, which lost its edge once the V^T build disappeared
  // and the 4-wave kernel fit 2 blocks/CU).  TORCHX_AMD_FWD_8WAVE=1
  // forces the 8-wave path for re-measurement.
  static int fwd8 = -1;
  if (fwd8 < 0) {
    const char* e = getenv("TORCHX_AMD_FWD_8WAVE");
    fwd8 = (e && atoi(e)) ? 1 : 0;
  }
  if (fwd8 == 1) {
    const int nqt = (S + 8 * QBLK - 1) / (8 * QBLK);
    hipLaunchKernelGGL((attn_fwd_kernel<8>), dim3(B * Hq * nqt), dim3(512),
                       0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (unsigned short*)o, (float*)lse, B, S, Hq, Hkv,
                       scale, causal, q_rs, kv_rs);
  } else {
    const int nqt = (S + 4 * QBLK - 1) / (4 * QBLK);
    hipLaunchKernelGGL((attn_fwd_kernel<4>), dim3(B * Hq * nqt), dim3(256),
                       0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (unsigned short*)o, (float*)lse, B, S, Hq, Hkv,
                       scale, causal, q_rs, kv_rs);
  }
}

extern "C" void attn_bwd_launch(const void* q, const void* k, const void* v,
                                const void* o, const void* dout,
                                const void* lse, void* delta, void* dq,
                                void* dk, void* dv, int B, int S, int Hq,
                                int Hkv, float scale, int causal,
                                long q_rs, long kv_rs, long dqkv_q_rs,
                                long dqkv_kv_rs, hipStream_t stream) {
  const long rows = (long)B * S * Hq;
  hipLaunchKernelGGL(attn_bwd_pre_kernel, dim3((int)((rows + 15) / 16)),
                     dim3(256), 0, stream, (const unsigned short*)dout,
                     (const unsigned short*)o, (float*)delta, rows, S, Hq);
  const int nqt = (S + BLOCK_Q - 1) / BLOCK_Q;
  static int dq_fkv = -1;
  if (dq_fkv < 0) {
    const char* e = getenv("TORCHX_AMD_DQ_FKV");
    dq_fkv = (e && atoi(e) == 128) ? 128 : 64;
  }
  if (dq_fkv == 128 && S % 128 == 0) {
    hipLaunchKernelGGL((attn_bwd_dq_kernel<128>), dim3(B * Hq * nqt),
                       dim3(256), 0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (const unsigned short*)dout, (const float*)lse,
                       (const float*)delta, (unsigned short*)dq, B, S, Hq,
                       Hkv, scale, causal, q_rs, kv_rs, dqkv_q_rs);
  } else {
    hipLaunchKernelGGL((attn_bwd_dq_kernel<64>), dim3(B * Hq * nqt),
                       dim3(256), 0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (const unsigned short*)dout, (const float*)lse,
                       (const float*)delta, (unsigned short*)dq, B, S, Hq,
                       Hkv, scale, causal, q_rs, kv_rs, dqkv_q_rs);
  }
  const int nkt = (S + BLOCK_K - 1) / BLOCK_K;
  const long dout_rs = (long)Hq * HD;
  // 128-row q tiles measured +4% over 64 (fewer barriers/iterations beat
  // the 2-blocks/CU the 64-row variant's smaller LDS allows)
  static int dkv_fkv = -1;
  if (dkv_fkv < 0) {
    const char* e = getenv("TORCHX_AMD_DKV_FKV");
    dkv_fkv = (e && atoi(e) == 64) ? 64 : 128;
  }
  if (dkv_fkv == 128 && S % 128 == 0) {
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<128>), dim3(B * Hkv * nkt),
                       dim3(512), 0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (const unsigned short*)dout, (const float*)lse,
                       (const float*)delta, (unsigned short*)dk,
                       (unsigned short*)dv, B, S, Hq, Hkv,
                       scale, causal, q_rs, kv_rs, dout_rs, dqkv_kv_rs);
  } else {
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<64>), dim3(B * Hkv * nkt),
                       dim3(512), 0, stream, (const unsigned short*)q,
                       (const unsigned short*)k, (const unsigned short*)v,
                       (const unsigned short*)dout, (const float*)lse,
                       (const float*)delta, (unsigned short*)dk,
                       (unsigned short*)dv, B, S, Hq, Hkv,
                       scale, causal, q_rs, kv_rs, dout_rs, dqkv_kv_rs);
  }
}

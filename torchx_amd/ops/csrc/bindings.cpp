// Torch extension bindings for the torchx_amd CDNA4 kernels.
// Compiled with hipcc for gfx950; no CUDA paths, no multi-backend dispatch.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <hip/hip_runtime.h>

extern "C" {
void rope_launch(const void*, void*, const void*, const void*, long, int, int,
                 int, float, hipStream_t);
void swiglu_fwd_launch(const void*, const void*, void*, long, hipStream_t);
void swiglu_bwd_launch(const void*, const void*, const void*, void*, void*,
                       long, hipStream_t);
void rmsnorm_fwd_launch(const void*, const void*, void*, void*, long, int,
                        float, hipStream_t);
void rmsnorm_bwd_launch(const void*, const void*, const void*, const void*,
                        void*, void*, long, int, hipStream_t);
void adamw_launch(void*, void*, const void*, void*, void*, long, float, float,
                  float, float, float, float, float, hipStream_t);
void ce_fwd_launch(const void*, const void*, void*, void*, long, int,
                   hipStream_t);
void ce_bwd_launch(const void*, const void*, const void*, const void*, void*,
                   long, int, hipStream_t);
void attn_fwd_launch(const void*, const void*, const void*, void*, void*, int,
                     int, int, int, float, int, long, long, hipStream_t);
void attn_bwd_launch(const void*, const void*, const void*, const void*,
                     const void*, const void*, void*, void*, void*, void*,
                     int, int, int, int, float, int, long, long, long, long,
                     hipStream_t);
void rope_qkv_launch(const void*, void*, const void*, const void*, long, int,
                     long, int, float, hipStream_t);
void swiglu_gu_fwd_launch(const void*, void*, long, long, hipStream_t);
void swiglu_gu_bwd_launch(const void*, const void*, void*, long, long,
                          hipStream_t);
void mfma_probe_launch(const void*, const void*, void*, hipStream_t);
void mfma_probe_tr_launch(const void*, const void*, void*, void*,
                          hipStream_t);
void fp8_cast_transpose_launch(const void*, void*, void*, void*,
                               const void*, long, long, int, hipStream_t);
void transpose_bf16_launch(const void*, void*, long, long, hipStream_t);
void decode_attn_launch(const void*, const void*, const void*, void*,
                        const void*, int, int, int, int, int, int, float,
                        hipStream_t);
void gemv_bf16_launch(const void*, const void*, void*, int, int, int,
                      hipStream_t);
void decode_attn_split_launch(const void*, const void*, const void*, void*,
                              void*, void*, const void*, int, int, int, int,
                              int, int, int, float, hipStream_t);
void decode_rope_cache_launch(const void*, void*, void*, void*, const void*,
                              const void*, const void*, int, int, int, int,
                              int, int, hipStream_t);
void rmsnorm_res_launch(const void*, const void*, const void*, void*, void*,
                        long, int, float, hipStream_t);
void gemv_swiglu_launch(const void*, const void*, void*, int, int, int,
                        hipStream_t);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_f32(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kFloat, name, " must be f32");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

}  // namespace

// ---- RMSNorm --------------------------------------------------------------
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const long H = x.size(-1);
  TORCH_CHECK(H % 2048 == 0 && H <= 8192, "rmsnorm: H must be k*2048, <=8192");
  const long rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto invrms = at::empty({rows}, x.options().dtype(at::kFloat));
  rmsnorm_fwd_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     invrms.data_ptr(), rows, (int)H, (float)eps,
                     cur_stream());
  return {y, invrms};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor invrms) {
  check_bf16(dy, "dy");
  check_bf16(x, "x");
  const long H = x.size(-1);
  const long rows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  rmsnorm_bwd_launch(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     invrms.data_ptr(), dx.data_ptr(), dw.data_ptr(), rows,
                     (int)H, cur_stream());
  return {dx, dw};
}

// ---- RoPE -----------------------------------------------------------------
at::Tensor rope(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, long heads,
                double sign) {
  check_bf16(x, "x");
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  const long D = x.size(-1);
  const long rows = x.numel() / D;
  const long S = cos_t.size(0);
  TORCH_CHECK(cos_t.size(1) == D / 2, "cos table must be [S, D/2]");
  auto y = at::empty_like(x);
  rope_launch(x.data_ptr(), y.data_ptr(), cos_t.data_ptr(), sin_t.data_ptr(),
              rows, (int)D, (int)heads, (int)S, (float)sign, cur_stream());
  return y;
}

// ---- SwiGLU ---------------------------------------------------------------
at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u) {
  check_bf16(g, "g");
  check_bf16(u, "u");
  auto out = at::empty_like(g);
  swiglu_fwd_launch(g.data_ptr(), u.data_ptr(), out.data_ptr(), g.numel(),
                    cur_stream());
  return out;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dout, at::Tensor g,
                                   at::Tensor u) {
  check_bf16(dout, "dout");
  auto dg = at::empty_like(g);
  auto du = at::empty_like(u);
  swiglu_bwd_launch(dout.data_ptr(), g.data_ptr(), u.data_ptr(), dg.data_ptr(),
                    du.data_ptr(), g.numel(), cur_stream());
  return {dg, du};
}

// ---- AdamW ----------------------------------------------------------------
void adamw_step(at::Tensor p32, at::Tensor p16, at::Tensor g, at::Tensor m,
                at::Tensor v, double lr, double beta1, double beta2,
                double eps, double wd, long step) {
  check_f32(p32, "p32");
  check_bf16(p16, "p16");
  check_bf16(g, "g");
  const long n = p32.numel();
  TORCH_CHECK(n % 4 == 0, "adamw: n must be a multiple of 4");
  const float bc1 = 1.0f - powf((float)beta1, (float)step);
  const float bc2 = 1.0f - powf((float)beta2, (float)step);
  adamw_launch(p32.data_ptr(), p16.data_ptr(), g.data_ptr(), m.data_ptr(),
               v.data_ptr(), n, (float)lr, (float)beta1, (float)beta2,
               (float)eps, (float)wd, bc1, bc2, cur_stream());
}

// ---- Cross entropy --------------------------------------------------------
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets) {
  check_bf16(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == at::kLong, "targets must be int64");
  const long V = logits.size(-1);
  const long T = logits.numel() / V;
  auto loss = at::empty({T}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({T}, logits.options().dtype(at::kFloat));
  ce_fwd_launch(logits.data_ptr(), targets.data_ptr(), loss.data_ptr(),
                lse.data_ptr(), T, (int)V, cur_stream());
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                  at::Tensor gscale) {
  check_bf16(logits, "logits");
  check_f32(lse, "lse");
  check_f32(gscale, "gscale");
  const long V = logits.size(-1);
  const long T = logits.numel() / V;
  auto dlogits = at::empty_like(logits);
  ce_bwd_launch(logits.data_ptr(), targets.data_ptr(), lse.data_ptr(),
                gscale.data_ptr(), dlogits.data_ptr(), T, (int)V,
                cur_stream());
  return dlogits;
}

// ---- Attention ------------------------------------------------------------
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double scale, bool causal) {
  check_bf16(q, "q");
  check_bf16(k, "k");
  check_bf16(v, "v");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128, "attention: head_dim must be 128");
  TORCH_CHECK(Hq % Hkv == 0, "attention: Hq must be a multiple of Hkv");
  auto o = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  attn_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  lse.data_ptr(), B, S, Hq, Hkv, (float)scale, causal ? 1 : 0,
                  (long)Hq * 128, (long)Hkv * 128, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor dout, at::Tensor lse,
                                 double scale, bool causal) {
  check_bf16(dout, "dout");
  const int B = q.size(0), S = q.size(1), Hq = q.size(2);
  const int Hkv = k.size(2);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  attn_bwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  dout.data_ptr(), lse.data_ptr(), delta.data_ptr(),
                  dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), B, S, Hq, Hkv,
                  (float)scale, causal ? 1 : 0, (long)Hq * 128,
                  (long)Hkv * 128, (long)Hq * 128, (long)Hkv * 128,
                  cur_stream());
  return {dq, dk, dv};
}

// ---- fused qkv attention (strided views over the packed qkv buffer) ------
// qkv: [B, S, (Hq+2*Hkv)*128] bf16 from the fused qkv GEMM.  q starts at 0,
// k at Hq*128, v at (Hq+Hkv)*128; row stride = (Hq+2*Hkv)*128.  Saves the
// split/contiguous copies forward and the torch.cat backward.
std::vector<at::Tensor> attn_fwd_qkv(at::Tensor qkv, long Hq, long Hkv,
                                     double scale, bool causal) {
  check_bf16(qkv, "qkv");
  const int B = qkv.size(0), S = qkv.size(1);
  const long rs = (Hq + 2 * Hkv) * 128;
  TORCH_CHECK(qkv.size(2) == rs, "qkv last dim mismatch");
  auto o = at::empty({B, S, Hq, 128}, qkv.options());
  auto lse = at::empty({B, Hq, S}, qkv.options().dtype(at::kFloat));
  const unsigned short* base = (const unsigned short*)qkv.data_ptr();
  attn_fwd_launch(base, base + Hq * 128, base + (Hq + Hkv) * 128,
                  o.data_ptr(), lse.data_ptr(), B, S, (int)Hq, (int)Hkv,
                  (float)scale, causal ? 1 : 0, rs, rs, cur_stream());
  return {o, lse};
}

at::Tensor attn_bwd_qkv(at::Tensor qkv, at::Tensor o, at::Tensor dout,
                        at::Tensor lse, long Hq, long Hkv, double scale,
                        bool causal) {
  check_bf16(dout, "dout");
  const int B = qkv.size(0), S = qkv.size(1);
  const long rs = (Hq + 2 * Hkv) * 128;
  auto dqkv = at::empty_like(qkv);
  auto delta = at::empty({B, Hq, S}, qkv.options().dtype(at::kFloat));
  const unsigned short* base = (const unsigned short*)qkv.data_ptr();
  unsigned short* dbase = (unsigned short*)dqkv.data_ptr();
  attn_bwd_launch(base, base + Hq * 128, base + (Hq + Hkv) * 128,
                  o.data_ptr(), dout.data_ptr(), lse.data_ptr(),
                  delta.data_ptr(), dbase, dbase + Hq * 128,
                  dbase + (Hq + Hkv) * 128, B, S, (int)Hq, (int)Hkv,
                  (float)scale, causal ? 1 : 0, rs, rs, rs, rs,
                  cur_stream());
  return dqkv;
}

// RoPE over the packed qkv buffer's q+k heads (y may alias x for in-place
// gradient rotation; the v region is copied when y != x).
at::Tensor rope_qkv(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                    long Hq, long Hkv, double sign, bool inplace) {
  check_bf16(x, "x");
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  const int B = x.size(0), S = x.size(1);
  const long rs = (Hq + 2 * Hkv) * 128;
  TORCH_CHECK(x.size(2) == rs, "qkv last dim mismatch");
  at::Tensor y = inplace ? x : at::empty_like(x);
  rope_qkv_launch(x.data_ptr(), y.data_ptr(), cos_t.data_ptr(),
                  sin_t.data_ptr(), (long)B * S, (int)(Hq + Hkv), rs, S,
                  (float)sign, cur_stream());
  if (!inplace) {
    const long v0 = (Hq + Hkv) * 128;
    y.narrow(2, v0, Hkv * 128).copy_(x.narrow(2, v0, Hkv * 128));
  }
  return y;
}

// ---- packed swiglu --------------------------------------------------------
at::Tensor swiglu_gu_fwd(at::Tensor gu) {
  check_bf16(gu, "gu");
  const long twoI = gu.size(-1);
  TORCH_CHECK(twoI % 16 == 0, "gu last dim must be 2I, I % 8 == 0");
  const long I = twoI / 2;
  const long rows = gu.numel() / twoI;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = at::empty(sizes, gu.options());
  swiglu_gu_fwd_launch(gu.data_ptr(), out.data_ptr(), rows, I, cur_stream());
  return out;
}

at::Tensor swiglu_gu_bwd(at::Tensor dout, at::Tensor gu) {
  check_bf16(dout, "dout");
  const long twoI = gu.size(-1);
  const long I = twoI / 2;
  const long rows = gu.numel() / twoI;
  auto dgu = at::empty_like(gu);
  swiglu_gu_bwd_launch(dout.data_ptr(), gu.data_ptr(), dgu.data_ptr(), rows,
                       I, cur_stream());
  return dgu;
}

// ---- fp8 fused cast+transpose ---------------------------------------------
// x bf16 [R, C] -> (fp8 [R, C], fp8 [C, R], amax_next scalar).
// scale divides the values (delayed scaling: pass the PREVIOUS amax-derived
// scale; amax_next feeds the next call).
std::vector<at::Tensor> fp8_cast_transpose(at::Tensor x, at::Tensor scale,
                                           bool e5m2) {
  check_bf16(x, "x");
  check_f32(scale, "scale");
  const long C = x.size(-1);
  const long R = x.numel() / C;
  TORCH_CHECK(R % 8 == 0 && C % 8 == 0, "fp8 cast: dims must be 8-aligned");
  auto dt = e5m2 ? at::kFloat8_e5m2 : at::kFloat8_e4m3fn;
  auto out = at::empty({R, C}, x.options().dtype(dt));
  auto out_t = at::empty({C, R}, x.options().dtype(dt));
  auto amax = at::zeros({1}, x.options().dtype(at::kFloat));
  fp8_cast_transpose_launch(x.data_ptr(), out.data_ptr(), out_t.data_ptr(),
                            amax.data_ptr(), scale.data_ptr(), R, C,
                            e5m2 ? 1 : 0, cur_stream());
  return {out, out_t, amax};
}

// ---- bf16 transpose -------------------------------------------------------
at::Tensor transpose_bf16(at::Tensor x) {
  check_bf16(x, "x");
  TORCH_CHECK(x.dim() == 2, "transpose_bf16: 2D only");
  const long R = x.size(0), C = x.size(1);
  TORCH_CHECK(R % 64 == 0 && C % 64 == 0,
              "transpose_bf16: dims must be multiples of 64");
  auto out = at::empty({C, R}, x.options());
  transpose_bf16_launch(x.data_ptr(), out.data_ptr(), R, C, cur_stream());
  return out;
}

// ---- decode attention -----------------------------------------------------
// q [B, Hq, 128] bf16; k/v caches [B, T, Hkv, 128] bf16; first L rows
// valid. Returns o [B, Hq, 128].
at::Tensor decode_attn(at::Tensor q, at::Tensor kc, at::Tensor vc, long L,
                       double scale) {
  check_bf16(q, "q");
  check_bf16(kc, "kc");
  check_bf16(vc, "vc");
  const int B = q.size(0), Hq = q.size(1);
  const int T = kc.size(1), Hkv = kc.size(2);
  TORCH_CHECK(q.size(2) == 128 && kc.size(3) == 128, "head_dim must be 128");
  TORCH_CHECK(L >= 1 && L <= T, "invalid cache length");
  TORCH_CHECK(Hq % Hkv == 0, "GQA group mismatch");
  auto o = at::empty_like(q);
  const int split = std::max(1, std::min(16, 512 / (B * Hq)));
  if (split > 1) {
    auto ml = at::empty({(long)B * Hq * split * 2},
                        q.options().dtype(at::kFloat));
    auto oacc = at::empty({(long)B * Hq * split * 128},
                          q.options().dtype(at::kFloat));
    decode_attn_split_launch(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                             ml.data_ptr(), oacc.data_ptr(), o.data_ptr(),
                             nullptr, B, Hq, Hkv, T, (int)L, split, 0,
                             (float)scale, cur_stream());
  } else {
    decode_attn_launch(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                       o.data_ptr(), nullptr, B, Hq, Hkv, T, (int)L, 0,
                       (float)scale, cur_stream());
  }
  return o;
}

// graph-capturable variant: cache length = pos + 1 read on DEVICE from an
// int32 scalar tensor, so a captured decode step stays valid as the cache
// grows between replays.
at::Tensor decode_attn_dev(at::Tensor q, at::Tensor kc, at::Tensor vc,
                           at::Tensor pos, double scale) {
  check_bf16(q, "q");
  const int B = q.size(0), Hq = q.size(1);
  TORCH_CHECK(pos.scalar_type() == at::kInt &&
                  (pos.numel() == 1 || pos.numel() == B),
              "pos must be an int32 scalar or [B] tensor on device");
  TORCH_CHECK(pos.is_contiguous(), "pos must be contiguous");
  const int per_row = pos.numel() == B && B > 1 ? 1 : 0;
  const int T = kc.size(1), Hkv = kc.size(2);
  auto o = at::empty_like(q);
  const int split = std::max(1, std::min(16, 512 / (B * Hq)));
  if (split > 1) {
    auto ml = at::empty({(long)B * Hq * split * 2},
                        q.options().dtype(at::kFloat));
    auto oacc = at::empty({(long)B * Hq * split * 128},
                          q.options().dtype(at::kFloat));
    decode_attn_split_launch(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                             ml.data_ptr(), oacc.data_ptr(), o.data_ptr(),
                             pos.data_ptr(), B, Hq, Hkv, T, 0, split,
                             per_row, (float)scale, cur_stream());
  } else {
    decode_attn_launch(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                       o.data_ptr(), pos.data_ptr(), B, Hq, Hkv, T, 0,
                       per_row, (float)scale, cur_stream());
  }
  return o;
}

// ---- skinny-M GEMV (decode linears) ---------------------------------------
// x [M, K] bf16 (M <= 8), w [N, K] bf16 (nn.Linear layout) -> out [M, N].
at::Tensor gemv_bf16(at::Tensor x, at::Tensor w) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "gemv: 2D only");
  const long M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(M >= 1 && M <= 8, "gemv: M must be 1..8");
  TORCH_CHECK(w.size(1) == K, "gemv: K mismatch");
  TORCH_CHECK(K % 512 == 0, "gemv: K must be a multiple of 512");
  auto out = at::empty({M, N}, x.options());
  gemv_bf16_launch(x.data_ptr(), w.data_ptr(), out.data_ptr(), (int)M,
                   (int)N, (int)K, cur_stream());
  return out;
}

// ---- fused skinny-M GEMV + SwiGLU (decode MLP) ----------------------------
// x [M, K], w [2I, K] packed (gate | up) -> silu(x@gate.T) * (x@up.T),
// [M, I]: the wgu projection and the swiglu activation in one dispatch.
at::Tensor gemv_swiglu_bf16(at::Tensor x, at::Tensor w) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "gemv_swiglu: 2D only");
  const long M = x.size(0), K = x.size(1), I = w.size(0) / 2;
  TORCH_CHECK(M >= 1 && M <= 8, "gemv_swiglu: M must be 1..8");
  TORCH_CHECK(w.size(0) == 2 * I && w.size(1) == K, "gemv_swiglu: shape");
  TORCH_CHECK(K % 512 == 0, "gemv_swiglu: K must be a multiple of 512");
  auto out = at::empty({M, I}, x.options());
  gemv_swiglu_launch(x.data_ptr(), w.data_ptr(), out.data_ptr(), (int)M,
                     (int)I, (int)K, cur_stream());
  return out;
}

// ---- fused decode rope + cache append -------------------------------------
// qkv [B, (Hq+2*Hkv)*128] (one position); ropes q/k with cos/sin [S, 64]
// f32 tables at row `pos` (host int, or device int32 scalar when `pos_dev`
// is given), appends k/v to the caches at `pos`, returns q [B, Hq, 128].
at::Tensor decode_rope_cache(at::Tensor qkv, at::Tensor kc, at::Tensor vc,
                             at::Tensor cos_t, at::Tensor sin_t, long pos,
                             c10::optional<at::Tensor> pos_dev) {
  check_bf16(qkv, "qkv");
  check_bf16(kc, "kc");
  check_bf16(vc, "vc");
  check_f32(cos_t, "cos");
  check_f32(sin_t, "sin");
  const int B = qkv.size(0), T = kc.size(1), Hkv = kc.size(2);
  TORCH_CHECK(kc.size(3) == 128 && cos_t.size(1) == 64,
              "head_dim must be 128");
  const long nh = qkv.size(1) / 128;
  const int Hq = (int)nh - 2 * Hkv;
  TORCH_CHECK(Hq >= 1 && qkv.size(1) == nh * 128, "qkv width mismatch");
  const void* pd = nullptr;
  int per_row = 0;
  if (pos_dev.has_value()) {
    TORCH_CHECK(pos_dev->scalar_type() == at::kInt &&
                    (pos_dev->numel() == 1 || pos_dev->numel() == B),
                "pos_dev must be an int32 scalar or [B] tensor");
    TORCH_CHECK(pos_dev->is_contiguous(), "pos_dev must be contiguous");
    pd = pos_dev->data_ptr();
    per_row = pos_dev->numel() == B && B > 1 ? 1 : 0;
  } else {
    TORCH_CHECK(pos >= 0 && pos < T, "pos out of cache range");
    TORCH_CHECK(pos < cos_t.size(0), "pos beyond rope table");
  }
  auto q = at::empty({B, (long)Hq, 128}, qkv.options());
  decode_rope_cache_launch(qkv.data_ptr(), q.data_ptr(), kc.data_ptr(),
                           vc.data_ptr(), cos_t.data_ptr(), sin_t.data_ptr(),
                           pd, B, Hq, Hkv, T, (int)pos, per_row,
                           cur_stream());
  return q;
}

// ---- fused residual add + rmsnorm (inference) ------------------------------
// s = x + res, y = rmsnorm(s) * w; returns (s, y). res omitted -> plain
// rmsnorm (s aliases x).
std::vector<at::Tensor> rmsnorm_res(at::Tensor x,
                                    c10::optional<at::Tensor> res,
                                    at::Tensor w, double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const long H = x.size(-1);
  TORCH_CHECK(H % 2048 == 0 && H <= 8192, "rmsnorm: H must be k*2048, <=8192");
  const long R = x.numel() / H;
  auto y = at::empty_like(x);
  at::Tensor s = x;
  const void* rp = nullptr;
  void* sp = nullptr;
  if (res.has_value()) {
    check_bf16(*res, "res");
    TORCH_CHECK(res->sizes() == x.sizes(), "res shape mismatch");
    s = at::empty_like(x);
    rp = res->data_ptr();
    sp = s.data_ptr();
  }
  rmsnorm_res_launch(x.data_ptr(), rp, w.data_ptr(), sp, y.data_ptr(), R,
                     (int)H, (float)eps, cur_stream());
  return {s, y};
}

// ---- probe ----------------------------------------------------------------
std::vector<at::Tensor> mfma_probe_tr(at::Tensor a, at::Tensor b) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  TORCH_CHECK(a.size(0) == 32 && a.size(1) == 16, "a must be [32,16]");
  TORCH_CHECK(b.size(0) == 16 && b.size(1) == 32, "b must be [16,32]");
  auto c = at::empty({32, 32}, a.options().dtype(at::kFloat));
  auto raw = at::empty({64, 8}, a.options());
  mfma_probe_tr_launch(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                       raw.data_ptr(), cur_stream());
  return {c, raw};
}

at::Tensor mfma_probe(at::Tensor a, at::Tensor b) {
  check_bf16(a, "a");
  check_bf16(b, "b");
  TORCH_CHECK(a.size(0) == 32 && a.size(1) == 16, "a must be [32,16]");
  TORCH_CHECK(b.size(0) == 16 && b.size(1) == 32, "b must be [16,32]");
  auto c = at::empty({32, 32}, a.options().dtype(at::kFloat));
  mfma_probe_launch(a.data_ptr(), b.data_ptr(), c.data_ptr(), cur_stream());
  return c;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rope", &rope);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("adamw_step", &adamw_step);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_fwd_qkv", &attn_fwd_qkv);
  m.def("attn_bwd_qkv", &attn_bwd_qkv);
  m.def("rope_qkv", &rope_qkv);
  m.def("swiglu_gu_fwd", &swiglu_gu_fwd);
  m.def("swiglu_gu_bwd", &swiglu_gu_bwd);
  m.def("fp8_cast_transpose", &fp8_cast_transpose);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma_probe_tr", &mfma_probe_tr);
  m.def("transpose_bf16", &transpose_bf16);
  m.def("decode_attn", &decode_attn);
  m.def("decode_attn_dev", &decode_attn_dev);
  m.def("gemv_bf16", &gemv_bf16);
  m.def("gemv_swiglu_bf16", &gemv_swiglu_bf16);
  m.def("decode_rope_cache", &decode_rope_cache,
        py::arg("qkv"), py::arg("kc"), py::arg("vc"), py::arg("cos"),
        py::arg("sin"), py::arg("pos") = 0,
        py::arg("pos_dev") = c10::nullopt);
  m.def("rmsnorm_res", &rmsnorm_res,
        py::arg("x"), py::arg("res"), py::arg("w"), py::arg("eps"));
}

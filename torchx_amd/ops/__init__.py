"""torchx_amd.ops — CDNA4 HIP hot-op library with autograd bindings.

On a GPU (MI355X) these ops REQUIRE the in-tree `_hip_ops.so` extension and
fail loudly if it is missing — there is no silent eager fallback on the GPU
path.  On CPU (CI containers have no GPU) they fall back to the fp32 torch
reference implementations in `torchx_amd.ops.reference`, which are also the
ground truth the GPU numerics tests compare against.
"""

from __future__ import annotations

import importlib.util
import math
from pathlib import Path
from typing import Optional, Tuple

import torch

from . import reference

_SO = Path(__file__).resolve().parent / "_hip_ops.so"
_hip = None
_load_err: Optional[str] = None


def _load():
    global _hip, _load_err
    if _hip is not None:
        return _hip
    if not _SO.exists():
        _load_err = f"{_SO} not built (run torchx_amd/ops/build.py)"
        return None
    try:
        # name must match TORCH_EXTENSION_NAME (PyInit__hip_ops)
        spec = importlib.util.spec_from_file_location("_hip_ops", _SO)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)  # type: ignore[union-attr]
        _hip = mod
    except Exception as e:  # noqa: BLE001
        _load_err = str(e)
    return _hip


def hip_ops(required: bool = True):
    """The raw extension module.  On a GPU box this must exist."""
    mod = _load()
    if mod is None and required:
        raise RuntimeError(
            f"torchx_amd HIP extension not available: {_load_err}. "
            "GPU execution requires the native kernels; refusing to fall "
            "back to eager."
        )
    return mod


def extension_available() -> bool:
    return _load() is not None


def _on_gpu(*ts: torch.Tensor) -> bool:
    return any(t.is_cuda for t in ts)


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, eps: float):
        y, invrms = hip_ops().rmsnorm_fwd(x.contiguous(), w.contiguous(), eps)
        ctx.save_for_backward(x, w, invrms)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w, invrms = ctx.saved_tensors
        dx, dw = hip_ops().rmsnorm_bwd(dy.contiguous(), x, w, invrms)
        return dx, dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _on_gpu(x):
        return _RMSNorm.apply(x, w, eps)
    return reference.rmsnorm(x, w, eps)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------


def rope_tables(
    seq_len: int, head_dim: int, theta: float = 500000.0,
    device: Optional[torch.device] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed f32 cos/sin tables [S, D/2] (guide: on-device trig
    turns a memory-bound elementwise op VALU-bound)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim)
    )
    pos = torch.arange(seq_len, dtype=torch.float32)
    ang = torch.outer(pos, inv_freq)  # [S, D/2]
    cos, sin = torch.cos(ang), torch.sin(ang)
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


class _RoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                heads: int):
        ctx.save_for_backward(cos, sin)
        ctx.heads = heads
        return hip_ops().rope(x.contiguous(), cos, sin, heads, 1.0)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        cos, sin = ctx.saved_tensors
        dx = hip_ops().rope(dy.contiguous(), cos, sin, ctx.heads, -1.0)
        return dx, None, None, None


def rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [B, S, H, D] bf16; rotate-half (Llama) convention."""
    heads = x.shape[2]
    if _on_gpu(x):
        return _RoPE.apply(x, cos, sin, heads)
    return reference.rope(x, cos, sin)


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g: torch.Tensor, u: torch.Tensor):
        ctx.save_for_backward(g, u)
        return hip_ops().swiglu_fwd(g.contiguous(), u.contiguous())

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        g, u = ctx.saved_tensors
        dg, du = hip_ops().swiglu_bwd(dout.contiguous(), g, u)
        return dg, du


def swiglu(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    if _on_gpu(g):
        return _SwiGLU.apply(g, u)
    return reference.swiglu(g, u)


# ---------------------------------------------------------------------------
# Fused cross entropy (mean reduction over T tokens)
# ---------------------------------------------------------------------------


class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, targets: torch.Tensor):
        V = logits.shape[-1]
        logits2d = logits.contiguous().view(-1, V)
        loss, lse = hip_ops().ce_fwd(logits2d, targets.contiguous().view(-1))
        ctx.save_for_backward(logits2d, targets, lse)
        ctx.orig_shape = logits.shape
        return loss.mean()

    @staticmethod
    def backward(ctx, gout: torch.Tensor):
        logits2d, targets, lse = ctx.saved_tensors
        T = logits2d.shape[0]
        gscale = (gout.float() / T).expand(T).contiguous()
        dlogits = hip_ops().ce_bwd(logits2d, targets.view(-1), lse, gscale)
        return dlogits.view(ctx.orig_shape), None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    if _on_gpu(logits):
        return _CrossEntropy.apply(logits, targets)
    return reference.cross_entropy(logits, targets)


# ---------------------------------------------------------------------------
# Fused lm_head GEMM + cross entropy (chunked — the [T, V] logits are never
# fully materialized or saved).  The CE loss is the last op of the step, so
# its input grads are known in forward up to the scalar gout: each token
# chunk runs  logits GEMM -> ce_fwd -> ce_bwd -> dx GEMM -> dW accumulation
# back-to-back while the chunk's logits are hot, then the chunk buffers are
# recycled.  dW accumulates in fp32 (bf16 += across chunks would drift).
# Cuts the saved-tensor footprint by 2*T*V bf16 (8.4 GB at the bench shape)
# and keeps the GEMM sizes fat (chunk=4096 rows).
# ---------------------------------------------------------------------------


class _FusedLinearCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2: torch.Tensor, w: torch.Tensor,
                targets: torch.Tensor, chunk: int):
        hip = hip_ops()
        T, H = x2.shape
        dx = torch.empty_like(x2)
        single = T <= chunk
        # multi-chunk dW accumulates in fp32 (bf16 += across chunks drifts);
        # a single chunk writes the bf16 wgrad GEMM output directly
        dw_acc = None if single else torch.zeros_like(w, dtype=torch.float32)
        loss_sum = torch.zeros((), device=x2.device, dtype=torch.float32)
        inv_t = 1.0 / T
        gscale = torch.full((min(chunk, T),), inv_t, device=x2.device,
                            dtype=torch.float32)
        # dx via the transposed-weight operand order (same ~15% dgrad win
        # as fast_linear); w transposed once for all chunks
        wt = (hip.transpose_bf16(w)
              if w.shape[0] % 64 == 0 and w.shape[1] % 64 == 0 else None)
        for s in range(0, T, chunk):
            e = min(T, s + chunk)
            x_c = x2[s:e]
            t_c = targets[s:e].contiguous()
            logits_c = x_c @ w.t()
            loss_c, lse_c = hip.ce_fwd(logits_c, t_c)
            loss_sum += loss_c.sum()
            dlog_c = hip.ce_bwd(logits_c, t_c, lse_c, gscale[:e - s])
            if wt is not None:
                torch.matmul(dlog_c, wt.t(), out=dx[s:e])
            else:
                torch.matmul(dlog_c, w, out=dx[s:e])
            if single:
                dw_acc = dlog_c.t() @ x_c
            else:
                dw_acc += dlog_c.t() @ x_c
        ctx.save_for_backward(dx, dw_acc)
        return loss_sum * inv_t

    @staticmethod
    def backward(ctx, gout: torch.Tensor):
        dx, dw_acc = ctx.saved_tensors
        # single scale(+cast) pass for dW; dx scales in place
        if dw_acc.dtype == dx.dtype:
            dw = dw_acc * gout.to(dx.dtype)
        else:
            dw = (dw_acc * gout.to(torch.float32)).to(dx.dtype)
        dx = dx * gout.to(dx.dtype)
        return dx, dw, None, None


def fused_linear_cross_entropy(
    x: torch.Tensor, w: torch.Tensor, targets: torch.Tensor,
    chunk: int = 16384,
) -> torch.Tensor:
    """mean CE of ``x @ w.T`` against ``targets`` without materializing the
    full logits. x [..., H] bf16, w [V, H] bf16, targets [...] int64."""
    H = x.shape[-1]
    x2 = x.reshape(-1, H)
    t = targets.reshape(-1)
    if _on_gpu(x):
        if torch.is_grad_enabled() and (x.requires_grad or w.requires_grad):
            return _FusedLinearCE.apply(x2.contiguous(), w, t, chunk)
        # eval path: loss only, no dx/dW work
        hip = hip_ops()
        x2 = x2.contiguous()
        T = x2.shape[0]
        loss_sum = torch.zeros((), device=x.device, dtype=torch.float32)
        for s in range(0, T, chunk):
            e = min(T, s + chunk)
            loss_c, _ = hip.ce_fwd(x2[s:e] @ w.t(), t[s:e].contiguous())
            loss_sum += loss_c.sum()
        return loss_sum / T
    return reference.cross_entropy(x2 @ w.t(), t)


# ---------------------------------------------------------------------------
# Flash attention (causal, GQA), BSHD layout
# ---------------------------------------------------------------------------


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                causal: bool, scale: float):
        o, lse = hip_ops().attn_fwd(
            q.contiguous(), k.contiguous(), v.contiguous(), scale, causal
        )
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = hip_ops().attn_bwd(
            q, k, v, o, dout.contiguous(), lse, ctx.scale, ctx.causal
        )
        return dq, dk, dv, None, None


def flash_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    causal: bool = True, scale: Optional[float] = None,
) -> torch.Tensor:
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 -> o [B,S,Hq,D]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        return _FlashAttention.apply(q, k, v, causal, scale)
    return reference.attention(q, k, v, causal=causal, scale=scale)


# ---------------------------------------------------------------------------
# Fused AdamW (flat-buffer optimizer step; see torchx_amd.parallel.optim)
# ---------------------------------------------------------------------------


def adamw_step(
    p32: torch.Tensor, p16: torch.Tensor, grad: torch.Tensor,
    m: torch.Tensor, v: torch.Tensor, *, lr: float, beta1: float = 0.9,
    beta2: float = 0.95, eps: float = 1e-8, weight_decay: float = 0.1,
    step: int = 1,
) -> None:
    if p32.is_cuda:
        hip_ops().adamw_step(p32, p16, grad, m, v, lr, beta1, beta2, eps,
                             weight_decay, step)
    else:
        reference.adamw_step(p32, p16, grad, m, v, lr=lr, beta1=beta1,
                             beta2=beta2, eps=eps, weight_decay=weight_decay,
                             step=step)


def mfma_probe(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return hip_ops().mfma_probe(a, b)


# ---------------------------------------------------------------------------
# fast_linear: bias-free linear whose dgrad uses the transposed-weight
# operand order.  dy @ Wt^T (Wt = transpose_bf16(W)) measures ~15% faster
# than dy @ W on every llama GEMM shape (tools/gemm_probe.py dgrad-NT vs
# dgrad-NN, ~1.55 vs ~1.33 PF/s); the transpose itself is an LDS-tiled
# kernel at HBM rate (~0.06 ms for the largest weight), so the swap nets
# ~17 ms/step at the bench config.
# ---------------------------------------------------------------------------


class _FastLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor):
        ctx.save_for_backward(x, w)
        return x @ w.t()

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w = ctx.saved_tensors
        hip = hip_ops()
        wt = hip.transpose_bf16(w)                      # [K, N]
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy2 @ wt.t()).view_as(x)
        # wgrad: for narrow outputs (qkv/wo/down shapes) transposing dy
        # once and running the wgrad as a plain NN GEMM beats the TN form
        # by ~15% incl. the transpose (tools/gemm_probe.py wgrad-viaT);
        # fat outputs (gu, lm_head) are neutral and skip the transpose
        M, N = dy2.shape
        via_t = N <= 8192 and M % 64 == 0 and N % 64 == 0
        # direct accumulation: when the param already has a (flat-buffer)
        # grad slot, the wgrad GEMM accumulates straight into it with
        # beta=1 — no dw buffer write/read and no AccumulateGrad RMW pass.
        # Safe because each weight is consumed once per step; the DDP
        # bucket hook is notified manually (grad_ready).
        wg = w.grad if w.is_leaf else None
        if wg is not None and wg.is_contiguous():
            a = hip.transpose_bf16(dy2) if via_t else dy2.t()
            if getattr(w, "_wgrad_fresh", False):
                # first touch since zero_grad: overwrite (beta=0) — skips
                # both the zeroed-C read in the GEMM and staleness
                torch.matmul(a, x2, out=wg)
                w._wgrad_fresh = False
            else:
                wg.addmm_(a, x2)
            from torchx_amd.parallel.ddp import grad_ready

            grad_ready(w)
            return dx, None
        dw = (hip.transpose_bf16(dy2) @ x2) if via_t else (dy2.t() @ x2)
        return dx, dw


def fast_linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """``x @ w.T`` with the NT-dgrad backward; falls back to F.linear off
    the fast path (CPU, or dims not 64-aligned)."""
    if (x.is_cuda and w.dtype == torch.bfloat16
            and w.shape[0] % 64 == 0 and w.shape[1] % 64 == 0):
        return _FastLinear.apply(x, w)
    return torch.nn.functional.linear(x, w)


# ---------------------------------------------------------------------------
# Fused qkv attention: rope(q,k) + flash attention straight off the packed
# qkv GEMM output — no split/contiguous copies forward, no torch.cat
# backward (the grads dq/dk/dv are written strided into one dqkv buffer and
# inverse-roped in place).
# ---------------------------------------------------------------------------


class _FusedAttentionQKV(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                Hq: int, Hkv: int, causal: bool, scale: float):
        hip = hip_ops()
        roped = hip.rope_qkv(qkv.contiguous(), cos, sin, Hq, Hkv, 1.0, False)
        o, lse = hip.attn_fwd_qkv(roped, Hq, Hkv, scale, causal)
        ctx.save_for_backward(roped, o, lse, cos, sin)
        ctx.dims = (Hq, Hkv, causal, scale)
        return o

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        roped, o, lse, cos, sin = ctx.saved_tensors
        Hq, Hkv, causal, scale = ctx.dims
        hip = hip_ops()
        dqkv = hip.attn_bwd_qkv(roped, o, dout.contiguous(), lse, Hq, Hkv,
                                scale, causal)
        # inverse rotation on the dq/dk regions, in place on our own buffer
        hip.rope_qkv(dqkv, cos, sin, Hq, Hkv, -1.0, True)
        return dqkv, None, None, None, None, None, None


def fused_attention_qkv(
    qkv: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
    Hq: int, Hkv: int, causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """qkv [B, S, (Hq+2*Hkv)*128] bf16 -> o [B, S, Hq, 128]."""
    D = qkv.shape[-1] // (Hq + 2 * Hkv)
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if _on_gpu(qkv):
        return _FusedAttentionQKV.apply(qkv, cos, sin, Hq, Hkv, causal, scale)
    # CPU reference path: split + rope + attention
    B, S, _ = qkv.shape
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = reference.rope(q.reshape(B, S, Hq, D), cos, sin)
    k = reference.rope(k.reshape(B, S, Hkv, D), cos, sin)
    return reference.attention(q, k, v.reshape(B, S, Hkv, D),
                               causal=causal, scale=scale)


class _SwiGLUPacked(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu: torch.Tensor):
        gu = gu.contiguous()
        ctx.save_for_backward(gu)
        return hip_ops().swiglu_gu_fwd(gu)

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        (gu,) = ctx.saved_tensors
        return hip_ops().swiglu_gu_bwd(dout.contiguous(), gu)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
    """gu [..., 2I] (gate | up) -> silu(gate) * up, [..., I]."""
    if _on_gpu(gu):
        return _SwiGLUPacked.apply(gu)
    g, u = gu.chunk(2, dim=-1)
    return reference.swiglu(g, u)


# ---------------------------------------------------------------------------
# Decode attention (serving): single-position attention over a KV cache —
# ops/csrc/decode_attn.hip, flash-decode style (wave-local online softmax,
# one merge barrier). No training/backward path: inference only.
# ---------------------------------------------------------------------------


def decode_attention(q: torch.Tensor, kcache: torch.Tensor,
                     vcache: torch.Tensor, length: int,
                     scale: Optional[float] = None) -> torch.Tensor:
    """q [B, Hq, 128] bf16; k/v caches [B, T, Hkv, 128] bf16 with the
    first ``length`` positions valid -> o [B, Hq, 128]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        return hip_ops().decode_attn(q.contiguous(), kcache, vcache,
                                     length, scale)
    return reference.decode_attention(q, kcache, vcache, length, scale)


def rmsnorm_res(x: torch.Tensor, res: Optional[torch.Tensor],
                w: torch.Tensor, eps: float = 1e-5):
    """Fused residual-add + RMSNorm for the decode path: returns
    ``(s, y)`` with ``s = x + res`` (the new residual stream, ``x`` when
    ``res`` is None) and ``y = rmsnorm(s) * w``. Inference only."""
    if _on_gpu(x):
        s, y = hip_ops().rmsnorm_res(x.contiguous(),
                                     res.contiguous() if res is not None
                                     else None, w.contiguous(), eps)
        return s, y
    s = x if res is None else (x + res)
    return s, reference.rmsnorm(s, w, eps)


def decode_rope_cache(qkv: torch.Tensor, kcache: torch.Tensor,
                      vcache: torch.Tensor, cos: torch.Tensor,
                      sin: torch.Tensor, pos,
                      num_heads: int) -> torch.Tensor:
    """Fused decode rope + KV-cache append: qkv [B, (Hq+2*Hkv)*D] for one
    position -> roped q [B, Hq, D]; ropes k and appends k/v to the caches
    at row ``pos`` (host int, or int32 device scalar for hipGraph capture).
    cos/sin are the FULL f32 [S, D/2] tables."""
    B = qkv.shape[0]
    Hkv, D = kcache.shape[2], kcache.shape[3]
    if _on_gpu(qkv):
        if isinstance(pos, torch.Tensor):
            return hip_ops().decode_rope_cache(qkv.contiguous(), kcache,
                                               vcache, cos, sin, 0, pos)
        return hip_ops().decode_rope_cache(qkv.contiguous(), kcache, vcache,
                                           cos, sin, int(pos), None)
    q, k, v = qkv.split([num_heads * D, Hkv * D, Hkv * D], dim=-1)
    q = q.reshape(B, 1, num_heads, D)
    k = k.reshape(B, 1, Hkv, D)
    v = v.reshape(B, 1, Hkv, D)
    if isinstance(pos, torch.Tensor) and pos.numel() > 1:
        outs = []  # ragged: each row ropes/appends at its own position
        for b in range(B):
            p = int(pos[b].item())
            cs, sn = cos[p:p + 1], sin[p:p + 1]
            outs.append(rope(q[b:b + 1].contiguous(), cs, sn))
            kcache[b, p:p + 1] = rope(k[b:b + 1].contiguous(), cs, sn)[0]
            vcache[b, p:p + 1] = v[b]
        return torch.cat(outs, dim=0).reshape(B, num_heads, D)
    p = int(pos.item()) if isinstance(pos, torch.Tensor) else int(pos)
    cs, sn = cos[p:p + 1], sin[p:p + 1]
    q = rope(q.contiguous(), cs, sn)
    k = rope(k.contiguous(), cs, sn)
    kcache[:, p:p + 1] = k
    vcache[:, p:p + 1] = v
    return q.reshape(B, num_heads, D)


def decode_linear_swiglu(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Fused decode MLP front half: ``silu(x @ gate.T) * (x @ up.T)`` for
    the packed wgu weight ``w = [gate; up]`` [2I, K] — one GEMV-shaped
    dispatch, no [M, 2I] intermediate. Inference only."""
    x2 = x.reshape(-1, x.shape[-1])
    M, K = x2.shape
    if _on_gpu(x) and M <= 8 and K % 512 == 0 and w.shape[0] % 2 == 0 \
            and w.dtype == torch.bfloat16 and w.is_contiguous():
        out = hip_ops().gemv_swiglu_bf16(x2.contiguous(), w)
    else:
        gu = x2 @ w.t()
        g, u = gu.chunk(2, dim=-1)
        out = torch.nn.functional.silu(g.float()).to(u.dtype) * u
    return out.reshape(*x.shape[:-1], w.shape[0] // 2)


def decode_attention_dev(q: torch.Tensor, kcache: torch.Tensor,
                         vcache: torch.Tensor, pos: torch.Tensor,
                         scale: Optional[float] = None) -> torch.Tensor:
    """Decode attention with the cache position(s) read from a tensor:
    ``pos`` is int32, a scalar (all rows at the same position — the
    hipGraph path) or [B] (ragged / continuous batching — each sequence
    attends over its own ``pos[b] + 1`` cache rows)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        return hip_ops().decode_attn_dev(q.contiguous(), kcache, vcache,
                                         pos.contiguous(), scale)
    if pos.numel() == 1:
        return reference.decode_attention(q, kcache, vcache,
                                          int(pos.item()) + 1, scale)
    outs = [reference.decode_attention(q[b:b + 1], kcache[b:b + 1],
                                       vcache[b:b + 1],
                                       int(pos[b].item()) + 1, scale)
            for b in range(q.shape[0])]
    return torch.cat(outs, dim=0)


def decode_linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Skinny-M linear for the decode path: x [..., M, K] with M <= 8
    total rows -> x @ w.T via the W-stream-bound ``gemv_bf16`` kernel
    (ops/csrc/gemv.hip). hipBLASLt's tile kernels reach only ~1/3 of the
    HBM roof at decode batch sizes; this kernel reads each weight byte
    exactly once. Falls back to torch.matmul off-GPU or when the shape
    doesn't qualify (M > 8 or K % 512 != 0). Inference only (no autograd).
    """
    x2 = x.reshape(-1, x.shape[-1])
    M, K = x2.shape
    # N > 32k (lm_head-sized): hipBLASLt's split-K kernels stream W
    # slightly faster there (5.8 vs 5.0 TB/s measured) — route to matmul
    if _on_gpu(x) and M <= 8 and K % 512 == 0 and w.shape[0] <= 32768 \
            and w.dtype == torch.bfloat16 and w.is_contiguous():
        out = hip_ops().gemv_bf16(x2.contiguous(), w)
    else:
        out = x2 @ w.t()
    return out.reshape(*x.shape[:-1], w.shape[0])

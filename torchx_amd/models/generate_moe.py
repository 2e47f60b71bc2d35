"""Serving-side generation for the Mixtral (MoE) family.

Same KV-cache prefill/decode shape as ``models/generate.py`` — the
attention half of a Mixtral block is identical to Llama, so it runs the
same fused decode kernels (skinny-M GEMV, decode_rope_cache, split-K
flash-decode attention, rmsnorm_res). The MLP half is top-k expert
routing: at decode batch B every (token, expert) group runs the fused
GEMV+SwiGLU expert FFN on its tokens only, so each step streams just the
ACTIVE experts' weights (<= B*top_k of num_experts per layer).

Single-rank serving only (``ep_size == 1`` — all experts local); the
expert-parallel all-to-all path in models/mixtral.py is a training
construct. Eager-only: MoE routing is data-dependent, so a hipGraph
capture would freeze one routing decision — the dense-model
``GraphedDecoder`` has no MoE counterpart by design.
"""

from __future__ import annotations

from typing import Optional

import torch

from torchx_amd import ops

from .generate import KVCache
from .mixtral import MixtralModel, MoELayer

__all__ = ["prefill_moe", "decode_step_moe", "generate_moe", "KVCache"]


def _moe_mlp_decode(moe: MoELayer, x: torch.Tensor) -> torch.Tensor:
    """Top-k MoE for a decode micro-batch x [B, H] -> [B, H]; experts run
    the skinny-M GEMV + fused SwiGLU path on their assigned tokens."""
    assert moe.ep_size == 1, "serving path is single-rank (all experts local)"
    cfg = moe.cfg
    logits = (x @ moe.router.weight.t()).float()          # [B, E]
    w, idx = torch.topk(logits, cfg.top_k, dim=-1)
    w = torch.softmax(w, dim=-1).to(x.dtype)              # [B, k]
    idx_h = idx.cpu()  # ONE host sync per layer; partitioning is host-side
    out = torch.zeros_like(x)
    for e in range(cfg.num_experts):
        rows_h = (idx_h == e).any(-1).nonzero(as_tuple=True)[0]
        if rows_h.numel() == 0:
            continue
        rows = rows_h.to(x.device, non_blocking=True)
        exp = moe.local_experts[e]
        h = ops.decode_linear_swiglu(x[rows], exp.wgu.weight)
        ye = ops.decode_linear(h, exp.wdown.weight)
        coef = (w * (idx == e)).sum(-1)[rows]             # [n], on device
        out[rows] += ye * coef[:, None]
    return out


@torch.no_grad()
def prefill_moe(model: MixtralModel, tokens: torch.Tensor,
                caches: list) -> torch.Tensor:
    """Prompt pass filling per-layer KV caches; returns last-position
    logits [B, vocab]. Mirrors generate.prefill with the MoE MLP."""
    cfg = model.cfg
    B, S = tokens.shape
    cos = model.rope_cos[:S]
    sin = model.rope_sin[:S]
    x = model.embed(tokens)
    for blk, cache in zip(model.blocks, caches):
        xn = ops.rmsnorm(x, blk.attn_norm, cfg.rms_eps)
        qkv = blk.wqkv(xn)
        q, k, v = qkv.split([cfg.q_dim, cfg.kv_dim, cfg.kv_dim], dim=-1)
        q = ops.rope(q.reshape(B, S, cfg.num_heads, cfg.head_dim)
                     .contiguous(), cos, sin)
        k = ops.rope(k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
                     .contiguous(), cos, sin)
        v = v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim).contiguous()
        cache.k[:, :S] = k
        cache.v[:, :S] = v
        cache.length = S
        attn = ops.flash_attention(q, k, v, causal=True)
        x = x + blk.wo(attn.reshape(B, S, cfg.q_dim))
        xn = ops.rmsnorm(x, blk.mlp_norm, cfg.rms_eps)
        x = x + blk.moe(xn)
    x = ops.rmsnorm(x, model.final_norm, cfg.rms_eps)
    return model.lm_head(x[:, -1])


@torch.no_grad()
def decode_step_moe(model: MixtralModel, token: torch.Tensor, caches: list,
                    pos_dev=None) -> torch.Tensor:
    """One token [B, 1] -> next-position logits [B, vocab]. Attention
    half = the fused Llama decode kernels; MLP half = active-expert
    GEMV+SwiGLU. An int32 [B] ``pos_dev`` runs the step RAGGED (every
    sequence at its own cache position — continuous batching); the MoE
    half is position-independent, so it needs no change."""
    cfg = model.cfg
    B = token.shape[0]
    blocks = model.blocks
    pos = caches[0].length if pos_dev is None else pos_dev
    x = model.embed(token).reshape(B, -1)
    _, xn = ops.rmsnorm_res(x, None, blocks[0].attn_norm, cfg.rms_eps)
    for i, (blk, cache) in enumerate(zip(blocks, caches)):
        qkv = ops.decode_linear(xn, blk.wqkv.weight)
        q = ops.decode_rope_cache(qkv, cache.k, cache.v, model.rope_cos,
                                  model.rope_sin, pos, cfg.num_heads)
        if pos_dev is None:
            cache.length = pos + 1
            o = ops.decode_attention(q, cache.k, cache.v, pos + 1)
        else:
            o = ops.decode_attention_dev(q, cache.k, cache.v, pos_dev)
        a = ops.decode_linear(o.reshape(B, -1), blk.wo.weight)
        x, xn = ops.rmsnorm_res(x, a, blk.mlp_norm, cfg.rms_eps)
        m = _moe_mlp_decode(blk.moe, xn)
        w_next = (blocks[i + 1].attn_norm if i + 1 < len(blocks)
                  else model.final_norm)
        x, xn = ops.rmsnorm_res(x, m, w_next, cfg.rms_eps)
    return ops.decode_linear(xn, model.lm_head.weight)


@torch.no_grad()
def generate_moe(
    model: MixtralModel,
    tokens: torch.Tensor,
    max_new_tokens: int,
    temperature: float = 0.0,
    top_k: Optional[int] = None,
    max_len: Optional[int] = None,
) -> torch.Tensor:
    """Greedy (temperature=0) or top-k sampled continuation for Mixtral.
    tokens [B, S0] -> [B, S0 + max_new_tokens]."""
    cfg = model.cfg
    B, S0 = tokens.shape
    total = S0 + max_new_tokens
    max_len = max_len or total
    assert total <= cfg.max_seq_len and max_len >= total
    caches = [KVCache.empty(cfg, B, max_len, tokens.device)
              for _ in range(cfg.num_layers)]

    def pick(logits: torch.Tensor) -> torch.Tensor:
        if temperature <= 0:
            return logits.argmax(-1, keepdim=True)
        logits = logits / temperature
        if top_k:
            kth = torch.topk(logits, top_k, dim=-1).values[:, -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        return torch.multinomial(torch.softmax(logits.float(), -1), 1)

    out = [tokens]
    nxt = pick(prefill_moe(model, tokens, caches))
    out.append(nxt)
    for _ in range(max_new_tokens - 1):
        nxt = pick(decode_step_moe(model, nxt, caches))
        out.append(nxt)
    return torch.cat(out, dim=1)

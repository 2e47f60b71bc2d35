"""Serving-side generation for the Llama family: prefill + KV-cache
decode on the CDNA4 kernels.

The training path never materializes K/V (the fused qkv attention keeps
them inside the kernel), so generation runs its own per-layer loop over
the SAME module weights: prefill uses the rope + flash-attention kernels
and captures the roped K/V into the cache; every decode step runs the
single-position ``ops.decode_attention`` kernel (flash-decode style)
against the cache. Inference only — ``torch.no_grad`` throughout.

No analog exists in the reference (a launcher); this is the
"deployment and serving" side of the bundled MI355X reference app.
288 GB HBM3E fits very large caches: a Llama-3-8B KV cache is
B * S * 8 heads * 128 * 2 (k+v) * 2 bytes = 8 KB per token-row
(~1 GB at B=4, S=32k).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from torchx_amd import ops

from .llama import LlamaConfig, LlamaModel


@dataclass
class KVCache:
    k: torch.Tensor  # [B, T, Hkv, 128] bf16
    v: torch.Tensor
    length: int = 0

    @staticmethod
    def empty(cfg: LlamaConfig, batch: int, max_len: int,
              device: torch.device) -> "KVCache":
        shape = (batch, max_len, cfg.num_kv_heads, cfg.head_dim)
        return KVCache(
            k=torch.zeros(shape, dtype=torch.bfloat16, device=device),
            v=torch.zeros(shape, dtype=torch.bfloat16, device=device),
        )


def _split_qkv(qkv: torch.Tensor, cfg: LlamaConfig):
    B, S, _ = qkv.shape
    q, k, v = qkv.split([cfg.q_dim, cfg.kv_dim, cfg.kv_dim], dim=-1)
    return (q.reshape(B, S, cfg.num_heads, cfg.head_dim).contiguous(),
            k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim).contiguous(),
            v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim).contiguous())


@torch.no_grad()
def prefill(model: LlamaModel, tokens: torch.Tensor,
            caches: list) -> torch.Tensor:
    """Run the prompt through the model, filling per-layer KV caches.
    Returns the last position's logits [B, vocab]."""
    cfg = model.cfg
    B, S = tokens.shape
    cos = model.rope_cos[:S]
    sin = model.rope_sin[:S]
    x = model.embed(tokens)
    for blk, cache in zip(model.blocks, caches):
        xn = ops.rmsnorm(x, blk.attn_norm, cfg.rms_eps)
        q, k, v = _split_qkv(blk.wqkv(xn), cfg)
        q = ops.rope(q, cos, sin)
        k = ops.rope(k, cos, sin)
        cache.k[:, :S] = k
        cache.v[:, :S] = v
        cache.length = S
        attn = ops.flash_attention(q, k, v, causal=True)
        x = x + blk.wo(attn.reshape(B, S, cfg.q_dim))
        xn = ops.rmsnorm(x, blk.mlp_norm, cfg.rms_eps)
        x = x + blk.wdown(ops.swiglu_packed(blk.wgu(xn)))
    x = ops.rmsnorm(x, model.final_norm, cfg.rms_eps)
    return model.lm_head(x[:, -1])


@torch.no_grad()
def decode_step(model: LlamaModel, token: torch.Tensor,
                caches: list) -> torch.Tensor:
    """One token [B, 1] -> next-position logits [B, vocab], appending to
    the caches."""
    cfg = model.cfg
    B = token.shape[0]
    pos = caches[0].length
    cos = model.rope_cos[pos:pos + 1]
    sin = model.rope_sin[pos:pos + 1]
    x = model.embed(token)  # [B, 1, H]
    for blk, cache in zip(model.blocks, caches):
        xn = ops.rmsnorm(x, blk.attn_norm, cfg.rms_eps)
        q, k, v = _split_qkv(blk.wqkv(xn), cfg)
        q = ops.rope(q, cos, sin)       # tables sliced at pos -> index 0
        k = ops.rope(k, cos, sin)
        cache.k[:, pos:pos + 1] = k
        cache.v[:, pos:pos + 1] = v
        cache.length = pos + 1
        o = ops.decode_attention(q.reshape(B, cfg.num_heads, cfg.head_dim),
                                 cache.k, cache.v, cache.length)
        x = x + blk.wo(o.reshape(B, 1, cfg.q_dim))
        xn = ops.rmsnorm(x, blk.mlp_norm, cfg.rms_eps)
        x = x + blk.wdown(ops.swiglu_packed(blk.wgu(xn)))
    x = ops.rmsnorm(x, model.final_norm, cfg.rms_eps)
    return model.lm_head(x[:, -1])


@torch.no_grad()
def generate(
    model: LlamaModel,
    tokens: torch.Tensor,
    max_new_tokens: int,
    temperature: float = 0.0,
    top_k: Optional[int] = None,
    max_len: Optional[int] = None,
) -> torch.Tensor:
    """Greedy (temperature=0) or top-k sampled continuation.
    tokens [B, S0] -> [B, S0 + max_new_tokens]."""
    cfg = model.cfg
    B, S0 = tokens.shape
    total = S0 + max_new_tokens
    max_len = max_len or total
    assert total <= cfg.max_seq_len, (total, cfg.max_seq_len)
    assert max_len >= total
    device = tokens.device
    caches = [KVCache.empty(cfg, B, max_len, device)
              for _ in range(cfg.num_layers)]

    def pick(logits: torch.Tensor) -> torch.Tensor:
        if temperature <= 0:
            return logits.argmax(-1, keepdim=True)
        logits = logits / temperature
        if top_k:
            kth = torch.topk(logits, top_k, dim=-1).values[:, -1:]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        probs = torch.softmax(logits.float(), dim=-1)
        return torch.multinomial(probs, 1)

    out = [tokens]
    nxt = pick(prefill(model, tokens, caches))
    out.append(nxt)
    for _ in range(max_new_tokens - 1):
        nxt = pick(decode_step(model, nxt, caches))
        out.append(nxt)
    return torch.cat(out, dim=1)

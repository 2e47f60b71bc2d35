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
def decode_step(model: LlamaModel, token: torch.Tensor, caches: list,
                pos_dev: Optional[torch.Tensor] = None) -> torch.Tensor:
    """One token [B, 1] -> next-position logits [B, vocab], appending to
    the caches. Fully fused decode layer (8 dispatches on GPU): skinny-M
    GEMV x3 + fused GEMV-SwiGLU x1, rope+cache-append x1, split-K
    flash-decode attention x2, fused residual-add+rmsnorm x2. With ``pos_dev`` (int32 device
    scalar) the step is hipGraph-capturable: the cache position comes off
    the device and no host state is read; an int32 [B] ``pos_dev`` runs
    the step RAGGED — every sequence at its own cache position
    (continuous batching, see ContinuousBatcher)."""
    cfg = model.cfg
    B = token.shape[0]
    blocks = model.blocks
    pos = caches[0].length if pos_dev is None else pos_dev
    x = model.embed(token).reshape(B, -1)  # [B, H] residual stream
    _, xn = ops.rmsnorm_res(x, None, blocks[0].attn_norm, cfg.rms_eps)
    for i, (blk, cache) in enumerate(zip(blocks, caches)):
        qkv = ops.decode_linear(xn, blk.wqkv.weight)
        q = ops.decode_rope_cache(qkv, cache.k, cache.v, model.rope_cos,
                                  model.rope_sin, pos, cfg.num_heads)
        if pos_dev is None:
            cache.length = pos + 1
            o = ops.decode_attention(q, cache.k, cache.v, pos + 1)
        else:
            # scalar pos_dev: hipGraph loop; [B] pos_dev: ragged decode
            # (continuous batching) — each row attends its own length
            o = ops.decode_attention_dev(q, cache.k, cache.v, pos_dev)
        a = ops.decode_linear(o.reshape(B, -1), blk.wo.weight)
        x, xn = ops.rmsnorm_res(x, a, blk.mlp_norm, cfg.rms_eps)
        m = ops.decode_linear(ops.decode_linear_swiglu(xn, blk.wgu.weight),
                              blk.wdown.weight)
        w_next = (blocks[i + 1].attn_norm if i + 1 < len(blocks)
                  else model.final_norm)
        x, xn = ops.rmsnorm_res(x, m, w_next, cfg.rms_eps)
    return ops.decode_linear(xn, model.lm_head.weight)


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


class GraphedDecoder:
    """hipGraph-captured decode loop: the whole per-token step (embed,
    32 layers, lm_head, argmax, cache append, position bump) replays as
    ONE graph with zero host work (the fused eager step is ~260
    dispatches; un-fused it was ~450 and launch-bound). The cache
    length and rope position are driven by a device int32 scalar that
    the captured step increments itself, so one capture serves every
    subsequent token.

    Greedy-only (the argmax feeds back inside the graph).
    """

    def __init__(self, model: LlamaModel, caches: list, batch: int,
                 first_token: torch.Tensor, start_pos: int):
        self.model = model
        self.caches = caches
        cfg = model.cfg
        dev = next(model.parameters()).device
        self.tok = first_token.clone()                      # [B, 1] int64
        self.pos32 = torch.tensor([start_pos], dtype=torch.int32,
                                  device=dev)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        # the eager warmup executes one REAL step (its token is collected
        # in init_tokens); the capture only RECORDS — nothing runs and no
        # state advances during it
        self.init_tokens = []
        self._step_body()                      # eager warm (same code path)
        torch.cuda.synchronize()
        self.init_tokens.append(self.tok.clone())
        with torch.cuda.graph(self.graph):
            self._step_body()
        torch.cuda.synchronize()

    @torch.no_grad()
    def _step_body(self) -> None:
        logits = decode_step(self.model, self.tok, self.caches,
                             pos_dev=self.pos32)
        self.tok.copy_(logits.argmax(-1, keepdim=True))
        self.pos32.add_(1)

    def step(self) -> torch.Tensor:
        """Replay one decode step; returns the new token [B, 1]."""
        self.graph.replay()
        return self.tok


@torch.no_grad()
def generate_graphed(model: LlamaModel, tokens: torch.Tensor,
                     max_new_tokens: int,
                     max_len: Optional[int] = None) -> torch.Tensor:
    """Greedy generation with the hipGraph decode loop (GPU only)."""
    cfg = model.cfg
    B, S0 = tokens.shape
    total = S0 + max_new_tokens
    max_len = max_len or total
    assert total <= cfg.max_seq_len and max_len >= total
    dev = tokens.device
    caches = [KVCache.empty(cfg, B, max_len, dev)
              for _ in range(cfg.num_layers)]
    out = [tokens]
    nxt = prefill(model, tokens, caches).argmax(-1, keepdim=True)
    out.append(nxt)
    if max_new_tokens <= 2:
        for _ in range(max_new_tokens - 1):
            nxt = decode_step(model, nxt, caches).argmax(-1, keepdim=True)
            out.append(nxt)
        return torch.cat(out, dim=1)
    dec = GraphedDecoder(model, caches, B, nxt, start_pos=S0)
    out.extend(t for t in dec.init_tokens[:max_new_tokens - 1])
    remaining = max_new_tokens - 1 - len(dec.init_tokens)
    for _ in range(remaining):
        out.append(dec.step().clone())
    return torch.cat(out, dim=1)


class ContinuousBatcher:
    """Continuous batching over a fixed pool of cache rows: sequences of
    DIFFERENT lengths decode together in one ragged step (int32 [B]
    position vector drives rope, cache append and attention per row), and
    a finished row can be re-admitted with a new prompt while the others
    keep decoding — the serving pattern behind vLLM-style engines,
    without paging (288 GB HBM3E holds the whole [B, T] cache pool).

    Greedy decoding; rows are independent — admit() prefills ONE row's
    cache slice, step() advances every active row one token.
    """

    def __init__(self, model, max_batch: int, max_len: int,
                 prefill_fn=None, decode_fn=None):
        """``prefill_fn``/``decode_fn`` default to the dense Llama path;
        pass ``generate_moe.prefill_moe`` / ``decode_step_moe`` to run a
        Mixtral pool (the MoE step takes the same ``pos_dev`` vector)."""
        self.model = model
        self.cfg = model.cfg
        self._prefill = prefill_fn or prefill
        self._decode = decode_fn or decode_step
        assert max_len <= self.cfg.max_seq_len, "beyond the rope tables"
        dev = next(model.parameters()).device
        self.device = dev
        self.max_len = max_len
        self.caches = [KVCache.empty(self.cfg, max_batch, max_len, dev)
                       for _ in range(self.cfg.num_layers)]
        self.pos = torch.zeros(max_batch, dtype=torch.int32, device=dev)
        # host mirror of pos (deterministic: admit sets it, step adds 1) —
        # overflow checks never touch the device
        self.pos_host = [0] * max_batch
        self.tok = torch.zeros(max_batch, 1, dtype=torch.long, device=dev)
        self.active = [False] * max_batch

    def free_rows(self) -> list:
        return [i for i, a in enumerate(self.active) if not a]

    @torch.no_grad()
    def admit(self, row: int, prompt: torch.Tensor) -> torch.Tensor:
        """Prefill ``prompt`` [S0] into cache row ``row``; returns the
        first generated token (scalar tensor). The row then participates
        in every subsequent step()."""
        assert not self.active[row], f"row {row} is occupied"
        S0 = prompt.numel()
        assert S0 + 1 < self.max_len
        row_caches = [
            KVCache(k=c.k[row:row + 1], v=c.v[row:row + 1], length=0)
            for c in self.caches
        ]
        logits = self._prefill(self.model,
                               prompt.reshape(1, S0).to(self.device),
                               row_caches)
        nxt = logits.argmax(-1)
        self.tok[row, 0] = nxt[0]
        self.pos[row] = S0
        self.pos_host[row] = S0
        self.active[row] = True
        return nxt[0]

    def retire(self, row: int) -> None:
        self.active[row] = False

    @torch.no_grad()
    def step(self) -> torch.Tensor:
        """One ragged decode step over ALL rows (inactive rows compute
        garbage that callers ignore — the batch shape stays static).
        Returns the new tokens [max_batch]."""
        assert any(self.active), "no active sequences"
        logits = self._decode(self.model, self.tok, self.caches,
                              pos_dev=self.pos)
        nxt = logits.argmax(-1, keepdim=True)
        self.tok.copy_(nxt)
        # bound every row (retired rows keep stepping as ignored garbage;
        # the wrap keeps their cache writes in range without a host sync)
        self.pos.add_(1).remainder_(self.max_len)
        for i in range(len(self.pos_host)):
            self.pos_host[i] = (self.pos_host[i] + 1) % self.max_len
            if self.active[i] and self.pos_host[i] + 1 >= self.max_len:
                self.active[i] = False  # out of cache: auto-retire
        return nxt.reshape(-1)

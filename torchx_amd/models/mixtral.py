"""Mixtral-8x7B reference model with expert parallelism over xGMI
(BASELINE.json config 5: dist.ddp 1x8 Mixtral 8x7B, expert-parallel RCCL
all-to-all).

Attention is identical to Llama (same CDNA4 HIP kernels); the MLP is a
top-2 MoE with 8 experts.  With ep_size ranks, each rank owns
E/ep_size experts; tokens are exchanged with all_to_all (parallel/ep.py);
dense (non-expert) parameters are data-parallel as usual — the DDP
all-reduce must therefore skip expert parameters when EP is active
(FlatParams puts experts in a separate group; see MixtralForEP notes).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from torchx_amd import ops
from torchx_amd.parallel.ep import exchange_counts, expert_all_to_all

from .llama import LlamaConfig, _lin


@dataclass
class MixtralConfig(LlamaConfig):
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    max_seq_len: int = 8192
    rope_theta: float = 1000000.0
    num_experts: int = 8
    top_k: int = 2


def mixtral_8x7b() -> MixtralConfig:
    return MixtralConfig()


def mixtral_tiny(vocab: int = 512) -> MixtralConfig:
    return MixtralConfig(
        vocab_size=vocab, hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
        max_seq_len=256, num_experts=4, top_k=2,
    )


def mixtral_gpu_tiny(vocab: int = 512) -> MixtralConfig:
    """Smallest MoE config on the HIP-kernel hot path (hidden 2048,
    head_dim 128 — see llama.llama_gpu_tiny)."""
    return MixtralConfig(
        vocab_size=vocab, hidden_size=2048, intermediate_size=2048,
        num_layers=2, num_heads=8, num_kv_heads=4, head_dim=128,
        max_seq_len=512, num_experts=4, top_k=2,
    )


class Expert(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.wgu = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                             bias=False, dtype=torch.bfloat16)
        self.wdown = nn.Linear(cfg.intermediate_size, cfg.hidden_size,
                               bias=False, dtype=torch.bfloat16)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _lin(self.wdown, ops.swiglu_packed(_lin(self.wgu, x)))


class MoELayer(nn.Module):
    """Top-k MoE with optional expert parallelism.

    ``ep_group`` (a torch.distributed group or None) shards the experts:
    rank r owns experts [r*E/ws, (r+1)*E/ws).  ep_group=None or world 1
    keeps all experts local (the CPU-testable path).
    """

    def __init__(self, cfg: MixtralConfig, ep_group=None,
                 ep_size: int = 1, ep_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        self.router = nn.Linear(cfg.hidden_size, cfg.num_experts, bias=False,
                                dtype=torch.bfloat16)
        self.ep_group = ep_group
        self.ep_size = ep_size
        self.ep_rank = ep_rank
        assert cfg.num_experts % ep_size == 0
        self.experts_per_rank = cfg.num_experts // ep_size
        self.local_experts = nn.ModuleList(
            [Expert(cfg) for _ in range(self.experts_per_rank)]
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, H = x.shape
        xt = x.reshape(-1, H)                                 # [T, H]
        T = xt.shape[0]
        logits = self.router(xt).float()                      # [T, E]
        weights, experts = torch.topk(logits, self.cfg.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(x.dtype)  # [T, k]

        k = self.cfg.top_k
        E = self.cfg.num_experts
        flat_expert = experts.reshape(-1)                     # [T*k]
        # sort token-copies by destination expert
        order = torch.argsort(flat_expert, stable=True)
        counts = torch.bincount(flat_expert, minlength=E)     # per expert
        xt_k = xt.repeat_interleave(k, dim=0)                 # [T*k, H]
        sorted_x = xt_k[order]

        if self.ep_size > 1:
            backend = dist.get_backend(self.ep_group)
            comm_dev = x.device if backend == "nccl" else torch.device("cpu")
            # per-rank send counts (sum over that rank's experts)
            counts_per_rank = counts.reshape(
                self.ep_size, self.experts_per_rank
            ).to(comm_dev)
            send = counts_per_rank.sum(-1)
            recv = exchange_counts(send, group=self.ep_group)
            in_splits = send.tolist()
            out_splits = recv.tolist()
            dispatched = expert_all_to_all(
                sorted_x, out_splits, in_splits, self.ep_group
            )
            # tokens arrive grouped by source rank, each group sorted by
            # LOCAL expert within the group; regroup by local expert across
            # sources: need every source's per-(rank, expert) counts
            all_counts = [torch.zeros_like(counts_per_rank)
                          for _ in range(self.ep_size)]
            dist.all_gather(all_counts, counts_per_rank, group=self.ep_group)
            my_counts = torch.stack(
                [c[self.ep_rank] for c in all_counts]
            ).cpu()  # [src_rank, experts_per_rank]
            # reorder dispatched tokens: currently [src0(e0..eN), src1(...)]
            # -> want grouped by local expert across sources
            segs = []
            offsets = my_counts.cumsum(dim=1) - my_counts  # start per (src,e)
            src_starts = torch.tensor(
                [0] + list(my_counts.sum(1).cumsum(0)[:-1])
            )
            for e in range(self.experts_per_rank):
                for src in range(self.ep_size):
                    start = int(src_starts[src] + offsets[src, e])
                    n = int(my_counts[src, e])
                    segs.append((e, src, start, n))
            perm = torch.cat(
                [torch.arange(s, s + n) for (_, _, s, n) in segs]
            ).to(dispatched.device)
            regrouped = dispatched[perm]
            expert_counts = my_counts.sum(0)                   # per local e
            outs = []
            off = 0
            for e in range(self.experts_per_rank):
                n = int(expert_counts[e])
                outs.append(self.local_experts[e](regrouped[off:off + n]))
                off += n
            computed = torch.cat(outs, dim=0) if outs else regrouped
            # inverse regroup
            inv_perm = torch.empty_like(perm)
            inv_perm[perm] = torch.arange(perm.numel(), device=perm.device)
            back = computed[inv_perm]
            returned = expert_all_to_all(
                back, in_splits, out_splits, self.ep_group
            )
        else:
            outs = []
            off = 0
            for e in range(E):
                n = int(counts[e])
                outs.append(self.local_experts[e](sorted_x[off:off + n]))
                off += n
            returned = torch.cat(outs, dim=0)

        # unpermute and weighted-combine the k copies (pure indexing keeps
        # autograd clean)
        inv_order = torch.argsort(order)
        unsorted = returned[inv_order]
        combined = (
            unsorted.reshape(T, k, H) * weights.unsqueeze(-1)
        ).sum(dim=1)
        return combined.reshape(B, S, H).to(x.dtype)


class MixtralBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig, ep_group=None, ep_size: int = 1,
                 ep_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        h = cfg.hidden_size
        self.wqkv = nn.Linear(h, cfg.q_dim + 2 * cfg.kv_dim, bias=False,
                              dtype=torch.bfloat16)
        self.wo = nn.Linear(cfg.q_dim, h, bias=False, dtype=torch.bfloat16)
        self.moe = MoELayer(cfg, ep_group, ep_size, ep_rank)
        self.attn_norm = nn.Parameter(torch.ones(h, dtype=torch.bfloat16))
        self.mlp_norm = nn.Parameter(torch.ones(h, dtype=torch.bfloat16))

    def forward(self, x, cos, sin):
        cfg = self.cfg
        B, S, H = x.shape
        xn = ops.rmsnorm(x, self.attn_norm, cfg.rms_eps)
        qkv = _lin(self.wqkv, xn)
        attn = ops.fused_attention_qkv(
            qkv, cos, sin, cfg.num_heads, cfg.num_kv_heads, causal=True
        )
        x = x + _lin(self.wo, attn.reshape(B, S, cfg.q_dim))
        xn = ops.rmsnorm(x, self.mlp_norm, cfg.rms_eps)
        return x + self.moe(xn)


class MixtralModel(nn.Module):
    def __init__(self, cfg: MixtralConfig,
                 device: Optional[torch.device] = None,
                 ep_group=None, ep_size: int = 1, ep_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                  dtype=torch.bfloat16)
        self.blocks = nn.ModuleList(
            [MixtralBlock(cfg, ep_group, ep_size, ep_rank)
             for _ in range(cfg.num_layers)]
        )
        self.final_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16)
        )
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False,
                                 dtype=torch.bfloat16)
        cos, sin = ops.rope_tables(cfg.max_seq_len, cfg.head_dim,
                                   cfg.rope_theta, device=device)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        if device is not None:
            self.to(device)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, tokens, targets=None):
        x = self.embed(tokens)
        for blk in self.blocks:
            x = blk(x, self.rope_cos, self.rope_sin)
        x = ops.rmsnorm(x, self.final_norm, self.cfg.rms_eps)
        if targets is None:
            return self.lm_head(x)
        return ops.fused_linear_cross_entropy(
            x, self.lm_head.weight, targets
        )

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())

"""Llama-3 family reference training model, MI355X-native.

This is the bundled reference training app the launcher is benchmarked on
(BASELINE.json: dist.ddp Llama-3-8B bf16).  Every hot op outside of GEMMs is
a hand-written CDNA4 HIP kernel from torchx_amd.ops (RMSNorm, RoPE, flash
attention, SwiGLU, fused cross-entropy); GEMMs run through hipBLASLt via
torch.nn.functional.linear.  Weights are bf16; the fused optimizer keeps
fp32 masters (torchx_amd.parallel.optim).

Layout conventions: activations [B, S, H*D]; attention tensors BSHD
(no transposes anywhere on the hot path — the attention kernel consumes
BSHD directly).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from torchx_amd import ops


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    tie_embeddings: bool = False

    @property
    def q_dim(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_dim(self) -> int:
        return self.num_kv_heads * self.head_dim


def llama3_8b() -> LlamaConfig:
    return LlamaConfig()


def llama_tiny(vocab: int = 512) -> LlamaConfig:
    """CPU-testable config."""
    return LlamaConfig(
        vocab_size=vocab, hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
        max_seq_len=256,
    )


def llama_gpu_tiny(vocab: int = 512) -> LlamaConfig:
    """Smallest config on the HIP-kernel hot path (the kernels specialize
    on production shapes: hidden multiple of 2048, head_dim 128)."""
    return LlamaConfig(
        vocab_size=vocab, hidden_size=2048, intermediate_size=4096,
        num_layers=2, num_heads=8, num_kv_heads=4, head_dim=128,
        max_seq_len=512,
    )


def _lin(mod: nn.Module, x: torch.Tensor) -> torch.Tensor:
    """Route plain bias-free nn.Linear through ops.fast_linear (dgrad via
    the transposed-weight operand, ~15% faster); anything else (Fp8Linear,
    CPU) goes through the module."""
    if type(mod) is nn.Linear and mod.bias is None and x.is_cuda:
        return ops.fast_linear(x, mod.weight)
    return mod(x)


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        h, qd, kvd = cfg.hidden_size, cfg.q_dim, cfg.kv_dim
        # fused qkv / gate-up projections: one GEMM each
        self.wqkv = nn.Linear(h, qd + 2 * kvd, bias=False, dtype=torch.bfloat16)
        self.wo = nn.Linear(qd, h, bias=False, dtype=torch.bfloat16)
        self.wgu = nn.Linear(h, 2 * cfg.intermediate_size, bias=False,
                             dtype=torch.bfloat16)
        self.wdown = nn.Linear(cfg.intermediate_size, h, bias=False,
                               dtype=torch.bfloat16)
        self.attn_norm = nn.Parameter(torch.ones(h, dtype=torch.bfloat16))
        self.mlp_norm = nn.Parameter(torch.ones(h, dtype=torch.bfloat16))

    def forward(self, x: torch.Tensor, cos: torch.Tensor,
                sin: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B, S, H = x.shape
        # attention: rope + flash straight off the packed qkv GEMM output
        # (no split/cat/contiguous traffic — ops.fused_attention_qkv)
        xn = ops.rmsnorm(x, self.attn_norm, cfg.rms_eps)
        qkv = _lin(self.wqkv, xn)
        attn = ops.fused_attention_qkv(
            qkv, cos, sin, cfg.num_heads, cfg.num_kv_heads, causal=True
        )
        x = x + _lin(self.wo, attn.reshape(B, S, cfg.q_dim))
        # mlp: swiglu over the packed gate|up buffer
        xn = ops.rmsnorm(x, self.mlp_norm, cfg.rms_eps)
        x = x + _lin(self.wdown, ops.swiglu_packed(_lin(self.wgu, xn)))
        return x


class LlamaModel(nn.Module):
    """Causal LM returning mean cross-entropy loss over shifted targets."""

    def __init__(self, cfg: LlamaConfig, device: Optional[torch.device] = None):
        super().__init__()
        self.cfg = cfg
        factory = {"dtype": torch.bfloat16, "device": device}
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size, **factory)
        self.blocks = nn.ModuleList(
            [LlamaBlock(cfg) for _ in range(cfg.num_layers)]
        )
        self.final_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16)
        )
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False,
                                 dtype=torch.bfloat16)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed.weight
        cos, sin = ops.rope_tables(cfg.max_seq_len, cfg.head_dim,
                                   cfg.rope_theta, device=device)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        if device is not None:
            self.to(device)
        self.apply(self._init_weights)

    def _init_weights(self, m: nn.Module) -> None:
        std = 0.02
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, mean=0.0, std=std)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, mean=0.0, std=std)

    def forward_hidden(self, tokens: torch.Tensor) -> torch.Tensor:
        x = self.embed(tokens)
        cos, sin = self.rope_cos, self.rope_sin
        for blk in self.blocks:
            x = blk(x, cos, sin)
        return ops.rmsnorm(x, self.final_norm, self.cfg.rms_eps)

    def forward(self, tokens: torch.Tensor,
                targets: Optional[torch.Tensor] = None) -> torch.Tensor:
        """tokens [B, S] -> loss (if targets given) else logits."""
        x = self.forward_hidden(tokens)
        if targets is None:
            return self.lm_head(x)
        # fused chunked lm_head GEMM + CE: the [T, vocab] logits are never
        # fully materialized or saved for backward (ops.fused_linear_cross_entropy)
        return ops.fused_linear_cross_entropy(
            x, self.lm_head.weight, targets
        )

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())

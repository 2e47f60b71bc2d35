"""Plain PyTorch fp32 reference implementations of every HIP hot op.

These are (a) the CPU execution path of `torchx_amd.ops` and (b) the ground
truth the GPU numerics tests compare the CDNA4 kernels against (the task's
test contract: HIP kernel vs plain PyTorch fp32 reference of the same op).
"""

from __future__ import annotations

import math
from typing import Optional

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    r = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * r * w.float()).to(x.dtype)


def rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x [B,S,H,D]; cos/sin [S,D/2]; rotate-half convention."""
    B, S, H, D = x.shape
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2:]
    c = cos[:S].view(1, S, 1, D // 2)
    s = sin[:S].view(1, S, 1, D // 2)
    y1 = x1 * c - x2 * s
    y2 = x2 * c + x1 * s
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def swiglu(g: torch.Tensor, u: torch.Tensor) -> torch.Tensor:
    gf = g.float()
    return (torch.nn.functional.silu(gf) * u.float()).to(g.dtype)


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    V = logits.shape[-1]
    return torch.nn.functional.cross_entropy(
        logits.float().view(-1, V), targets.view(-1)
    )


def attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    causal: bool = True, scale: Optional[float] = None,
) -> torch.Tensor:
    """BSHD GQA attention in fp32 math; returns bf16 like the kernel."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    g = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)                       # [B,Hq,S,D]
    kf = k.float().permute(0, 2, 1, 3)                       # [B,Hkv,S,D]
    vf = v.float().permute(0, 2, 1, 3)
    kf = kf.repeat_interleave(g, dim=1)
    vf = vf.repeat_interleave(g, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale       # [B,Hq,S,S]
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, vf)                                   # [B,Hq,S,D]
    return o.permute(0, 2, 1, 3).contiguous().to(q.dtype)


def attention_lse(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    causal: bool = True, scale: Optional[float] = None,
) -> torch.Tensor:
    """Reference log-sum-exp [B,Hq,S] matching the kernel's saved lse."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    g = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(g, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    return torch.logsumexp(s, dim=-1)


def adamw_step(
    p32: torch.Tensor, p16: torch.Tensor, grad: torch.Tensor,
    m: torch.Tensor, v: torch.Tensor, *, lr: float, beta1: float,
    beta2: float, eps: float, weight_decay: float, step: int,
) -> None:
    gf = grad.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    mhat = m / bc1
    denom = (v / bc2).sqrt().add_(eps)
    p32.add_(mhat / denom + weight_decay * p32, alpha=-lr)
    p16.copy_(p32.to(p16.dtype))


def decode_attention(q: torch.Tensor, kcache: torch.Tensor,
                     vcache: torch.Tensor, length: int,
                     scale: float) -> torch.Tensor:
    """fp32 reference for single-position attention over a KV cache.
    q [B, Hq, D]; k/v caches [B, T, Hkv, D]; first ``length`` rows valid."""
    B, Hq, D = q.shape
    Hkv = kcache.shape[2]
    G = Hq // Hkv
    k = kcache[:, :length].float()                      # [B, L, Hkv, D]
    v = vcache[:, :length].float()
    k = k.repeat_interleave(G, dim=2)                   # [B, L, Hq, D]
    v = v.repeat_interleave(G, dim=2)
    s = torch.einsum("bhd,blhd->bhl", q.float(), k) * scale
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bhl,blhd->bhd", p, v).to(q.dtype)

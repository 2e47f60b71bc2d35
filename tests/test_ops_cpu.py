"""CPU-fallback numerics for the fused ops: each public op's CPU path is
checked against an INDEPENDENT plain-torch formulation (the GPU kernels are
checked against the same formulations in test_ops_gpu.py — this pins the
fallback so the two paths can't drift apart silently)."""

import math

import pytest
import torch
import torch.nn.functional as F

from torchx_amd import ops

torch.manual_seed(0)


def test_rmsnorm_cpu_matches_manual():
    x = torch.randn(4, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y = ops.rmsnorm(x, w, eps=1e-5)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(y, ref, atol=1e-6)
    y.sum().backward()
    assert x.grad is not None and w.grad is not None


def test_rope_cpu_matches_manual_rotation():
    B, S, H, D = 2, 8, 3, 16
    x = torch.randn(B, S, H, D)
    cos, sin = ops.rope_tables(S, D)
    y = ops.rope(x, cos, sin)
    # independent rotate-half formulation
    x1, x2 = x[..., : D // 2], x[..., D // 2:]
    c = cos[:S].view(1, S, 1, D // 2)
    s = sin[:S].view(1, S, 1, D // 2)
    ref = torch.cat([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)
    assert torch.allclose(y, ref, atol=1e-6)


def test_swiglu_cpu_and_packed_agree():
    g = torch.randn(5, 32, requires_grad=True)
    u = torch.randn(5, 32, requires_grad=True)
    y = ops.swiglu(g, u)
    assert torch.allclose(y, F.silu(g) * u, atol=1e-6)
    gu = torch.cat([g, u], dim=-1)
    assert torch.allclose(ops.swiglu_packed(gu), y, atol=1e-6)
    # gradients flow through the packed path
    gu2 = torch.cat([g, u], dim=-1).detach().requires_grad_(True)
    ops.swiglu_packed(gu2).sum().backward()
    assert gu2.grad is not None and gu2.grad.abs().sum() > 0


def test_cross_entropy_cpu_matches_torch():
    logits = torch.randn(12, 50, requires_grad=True)
    targets = torch.randint(0, 50, (12,))
    loss = ops.cross_entropy(logits, targets)
    ref = F.cross_entropy(logits, targets)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    logits2 = logits.detach().requires_grad_(True)
    F.cross_entropy(logits2, targets).backward()
    assert torch.allclose(logits.grad, logits2.grad, atol=1e-6)


def test_flash_attention_cpu_matches_sdpa():
    B, S, Hq, Hkv, D = 1, 16, 4, 2, 32
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    o = ops.flash_attention(q, k, v, causal=True)
    # independent reference: torch SDPA with GQA expansion, SBHD->BHSD
    kq = k.repeat_interleave(Hq // Hkv, dim=2)
    vq = v.repeat_interleave(Hq // Hkv, dim=2)
    ref = F.scaled_dot_product_attention(
        q.transpose(1, 2), kq.transpose(1, 2), vq.transpose(1, 2),
        is_causal=True, scale=1.0 / math.sqrt(D),
    ).transpose(1, 2)
    assert torch.allclose(o, ref, atol=1e-5), (o - ref).abs().max()


def test_fused_attention_qkv_cpu_matches_unfused():
    B, S, Hq, Hkv, D = 1, 8, 4, 2, 16
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D)
    cos, sin = ops.rope_tables(S, D)
    o = ops.fused_attention_qkv(qkv, cos, sin, Hq, Hkv)
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = ops.rope(q.reshape(B, S, Hq, D), cos, sin)
    k = ops.rope(k.reshape(B, S, Hkv, D), cos, sin)
    ref = ops.flash_attention(q, k, v.reshape(B, S, Hkv, D), causal=True)
    assert torch.allclose(o, ref, atol=1e-6)


def test_adamw_step_cpu_matches_torch_optim():
    torch.manual_seed(1)
    n = 257
    p32 = torch.randn(n)
    grad = torch.randn(n)
    m = torch.zeros(n)
    v = torch.zeros(n)
    p16 = p32.to(torch.bfloat16)
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.95, 1e-8, 0.1

    ref_p = torch.nn.Parameter(p32.clone())
    opt = torch.optim.AdamW([ref_p], lr=lr, betas=(b1, b2), eps=eps,
                            weight_decay=wd)
    ref_p.grad = grad.clone()
    opt.step()

    ops.adamw_step(p32, p16, grad, m, v, lr=lr, beta1=b1, beta2=b2,
                   eps=eps, weight_decay=wd, step=1)
    assert torch.allclose(p32, ref_p.detach(), atol=1e-6), \
        (p32 - ref_p.detach()).abs().max()
    assert torch.allclose(p16, p32.to(torch.bfloat16))


def test_decode_linear_cpu_unit():
    import torch

    from torchx_amd import ops

    x = torch.randn(3, 64)
    w = torch.randn(10, 64)
    assert torch.allclose(ops.decode_linear(x, w), x @ w.t(), atol=1e-5)


def test_decode_linear_swiglu_cpu_unit():
    import torch

    from torchx_amd import ops

    x = torch.randn(2, 32)
    w = torch.randn(12, 32)  # [gate(6); up(6)]
    out = ops.decode_linear_swiglu(x, w)
    gu = x @ w.t()
    g, u = gu.chunk(2, -1)
    ref = torch.nn.functional.silu(g) * u
    assert out.shape == (2, 6)
    assert torch.allclose(out, ref, atol=1e-5)


def test_decode_attention_dev_cpu_ragged():
    import torch

    from torchx_amd import ops
    from torchx_amd.ops import reference

    torch.manual_seed(3)
    B, Hq, Hkv, T, D = 3, 4, 2, 10, 16
    q = torch.randn(B, Hq, D)
    kc = torch.randn(B, T, Hkv, D)
    vc = torch.randn(B, T, Hkv, D)
    pos = torch.tensor([2, 7, 4], dtype=torch.int32)
    out = ops.decode_attention_dev(q, kc, vc, pos)
    for b in range(B):
        ref = reference.decode_attention(
            q[b:b + 1], kc[b:b + 1], vc[b:b + 1], int(pos[b]) + 1,
            1.0 / D ** 0.5)
        assert torch.allclose(out[b:b + 1], ref, atol=1e-4)

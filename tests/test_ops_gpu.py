"""GPU numerics tests: every CDNA4 HIP kernel vs the plain PyTorch fp32
reference of the same op (tests/conftest.py registers the gpu marker)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    return torch.device("cuda:0")


def _hip():
    from torchx_amd import ops

    assert ops.extension_available(), "HIP extension must be built in-tree"
    return ops


def rel_err(a, b):
    a, b = a.float(), b.float()
    return (a - b).abs().max().item() / (b.abs().max().item() + 1e-6)


def test_mfma_layout_probe(dev):
    """Validates the assumed A/B/C fragment lane maps for
    v_mfma_f32_32x32x16_bf16 against torch.matmul (asymmetric inputs)."""
    ops = _hip()
    a = torch.randn(32, 16, device=dev).to(torch.bfloat16)
    b = torch.randn(16, 32, device=dev).to(torch.bfloat16)
    c = ops.mfma_probe(a, b)
    ref = a.float() @ b.float()
    assert torch.allclose(c, ref, atol=2e-2, rtol=2e-2), (
        f"MFMA layout mismatch: max err {(c - ref).abs().max().item()}"
    )


def test_rmsnorm_fwd_bwd(dev):
    ops = _hip()
    x = torch.randn(512, 4096, device=dev, dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(4096, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = ops.rmsnorm(x, w)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    yr = (xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5)) * wr
    assert rel_err(y, yr) < 2e-2

    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert rel_err(x.grad, xr.grad) < 3e-2
    assert rel_err(w.grad, wr.grad) < 3e-2


def test_rope_fwd_bwd(dev):
    from torchx_amd.ops import reference, rope_tables

    ops = _hip()
    B, S, H, D = 2, 256, 4, 128
    cos, sin = rope_tables(S, D, device=dev)
    x = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = ops.rope(x, cos, sin)
    yr = reference.rope(x.detach(), cos, sin)
    assert rel_err(y, yr) < 2e-2
    dy = torch.randn_like(y)
    y.backward(dy)
    # inverse-rotation backward equals reference autograd backward
    xr = x.detach().float().requires_grad_(True)
    c = cos[:S].view(1, S, 1, D // 2)
    s = sin[:S].view(1, S, 1, D // 2)
    x1, x2 = xr[..., : D // 2], xr[..., D // 2:]
    yref = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1)
    yref.backward(dy.float())
    assert rel_err(x.grad, xr.grad) < 2e-2


def test_swiglu_fwd_bwd(dev):
    ops = _hip()
    g = torch.randn(4096, 1024, device=dev, dtype=torch.bfloat16, requires_grad=True)
    u = torch.randn_like(g).requires_grad_(True)
    out = ops.swiglu(g, u)
    gr = g.detach().float().requires_grad_(True)
    ur = u.detach().float().requires_grad_(True)
    outr = torch.nn.functional.silu(gr) * ur
    assert rel_err(out, outr) < 2e-2
    do = torch.randn_like(out)
    out.backward(do)
    outr.backward(do.float())
    assert rel_err(g.grad, gr.grad) < 2e-2
    assert rel_err(u.grad, ur.grad) < 2e-2


def test_cross_entropy(dev):
    ops = _hip()
    T, V = 128, 128256
    logits = torch.randn(T, V, device=dev, dtype=torch.bfloat16, requires_grad=True)
    targets = torch.randint(0, V, (T,), device=dev)
    loss = ops.cross_entropy(logits, targets)
    lr_ = logits.detach().float().requires_grad_(True)
    loss_ref = torch.nn.functional.cross_entropy(lr_, targets)
    assert abs(loss.item() - loss_ref.item()) < 2e-2
    loss.backward()
    loss_ref.backward()
    assert rel_err(logits.grad, lr_.grad) < 3e-2


def test_transpose_bf16(dev):
    ops = _hip()
    hip = ops.hip_ops()
    for R, C in [(256, 128), (4096, 6144), (128, 128256 // 2)]:
        x = torch.randn(R, C, device=dev, dtype=torch.bfloat16)
        t = hip.transpose_bf16(x)
        assert t.shape == (C, R)
        assert torch.equal(t, x.t().contiguous())


def test_fast_linear(dev):
    ops = _hip()
    torch.manual_seed(5)
    x = torch.randn(4, 128, 1024, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(2048, 1024, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.fast_linear(x, w)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    yr = xr @ wr.t()
    assert rel_err(y, yr) < 2e-2
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert rel_err(x.grad, xr.grad) < 2e-2
    assert rel_err(w.grad, wr.grad) < 2e-2


def test_fast_linear_direct_grad_accumulate(dev):
    """With a pre-existing (flat-buffer-style) grad slot, fast_linear's
    wgrad accumulates in place via addmm_ and never materializes dw."""
    ops = _hip()
    torch.manual_seed(9)
    for N in (2048, 12288):  # via-transpose and TN paths
        x = torch.randn(256, 1024, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        w = torch.randn(N, 1024, device=dev, dtype=torch.bfloat16,
                        requires_grad=True) * 0.05
        w = w.detach().requires_grad_(True)
        seed_grad = torch.randn_like(w) * 0.01
        w.grad = seed_grad.clone()
        y = ops.fast_linear(x, w)
        dy = torch.randn_like(y)
        y.backward(dy)
        # reference: seed + dy^T @ x in fp32
        ref = seed_grad.float() + dy.float().t() @ x.detach().float()
        err = (w.grad.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err < 3e-2 * scale, (N, err, scale)
        assert x.grad is not None


def test_fused_linear_cross_entropy(dev):
    ops = _hip()
    T, H, V = 384, 1024, 4096
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(V, H, device=dev, dtype=torch.bfloat16,
                    requires_grad=True) * 0.05
    w = w.detach().requires_grad_(True)
    targets = torch.randint(0, V, (T,), device=dev)
    loss = ops.fused_linear_cross_entropy(x, w, targets, chunk=128)

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    loss_ref = torch.nn.functional.cross_entropy(xr @ wr.t(), targets)
    assert abs(loss.item() - loss_ref.item()) < 2e-2

    # non-unit upstream grad exercises the deferred gout scaling
    (loss * 3.0).backward()
    (loss_ref * 3.0).backward()
    assert rel_err(x.grad, xr.grad) < 3e-2
    assert rel_err(w.grad, wr.grad) < 3e-2


def test_fused_linear_ce_matches_unfused(dev):
    """Fused path vs the unfused hip lm_head+CE path on the same inputs."""
    ops = _hip()
    T, H, V = 256, 2048, 8192
    torch.manual_seed(3)
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    w = torch.randn(V, H, device=dev, dtype=torch.bfloat16) * 0.05
    targets = torch.randint(0, V, (T,), device=dev)

    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    loss1 = ops.fused_linear_cross_entropy(x1, w1, targets, chunk=64)
    loss1.backward()

    x2 = x.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    loss2 = ops.cross_entropy(x2 @ w2.t(), targets)
    loss2.backward()

    assert abs(loss1.item() - loss2.item()) < 1e-2
    assert rel_err(x1.grad, x2.grad) < 2e-2
    assert rel_err(w1.grad, w2.grad) < 2e-2


@pytest.mark.parametrize("S,causal", [(128, True), (256, True), (512, True),
                                      (192, True), (256, False)])
def test_attention_fwd(dev, S, causal):
    ops = _hip()
    B, Hq, Hkv, D = 2, 8, 2, 128
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    from torchx_amd.ops import reference

    o = ops.flash_attention(q, k, v, causal=causal)
    o_ref = reference.attention(q, k, v, causal=causal)
    assert rel_err(o, o_ref) < 3e-2, f"attention fwd err {rel_err(o, o_ref)}"


def test_attention_lse(dev):
    ops = _hip()
    from torchx_amd.ops import reference

    B, S, Hq, Hkv, D = 1, 256, 4, 4, 128
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    o, lse = ops.hip_ops().attn_fwd(q, k, v, 1.0 / math.sqrt(D), True)
    lse_ref = reference.attention_lse(q, k, v, causal=True)
    assert rel_err(lse, lse_ref) < 2e-2


@pytest.mark.parametrize("S,causal", [(256, True), (192, True), (128, False)])
def test_attention_bwd(dev, S, causal):
    ops = _hip()
    B, Hq, Hkv, D = 2, 8, 2, 128
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    o = ops.flash_attention(q, k, v, causal=causal)
    do = torch.randn_like(o)
    o.backward(do)

    from torchx_amd.ops import reference

    qr = q.detach().float().requires_grad_(True)
    kr = k.detach().float().requires_grad_(True)
    vr = v.detach().float().requires_grad_(True)
    g = Hq // Hkv
    qf = qr.permute(0, 2, 1, 3)
    kf = kr.permute(0, 2, 1, 3).repeat_interleave(g, dim=1)
    vf = vr.permute(0, 2, 1, 3).repeat_interleave(g, dim=1)
    s = qf @ kf.transpose(-1, -2) / math.sqrt(D)
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=dev).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    o_ref = (torch.softmax(s, -1) @ vf).permute(0, 2, 1, 3)
    o_ref.backward(do.float())
    assert rel_err(q.grad, qr.grad) < 4e-2, f"dq err {rel_err(q.grad, qr.grad)}"
    assert rel_err(k.grad, kr.grad) < 4e-2, f"dk err {rel_err(k.grad, kr.grad)}"
    assert rel_err(v.grad, vr.grad) < 4e-2, f"dv err {rel_err(v.grad, vr.grad)}"


def test_adamw(dev):
    ops = _hip()
    from torchx_amd.ops import reference

    n = 8192
    p32 = torch.randn(n, device=dev)
    p16 = p32.to(torch.bfloat16)
    g = torch.randn(n, device=dev, dtype=torch.bfloat16)
    m = torch.zeros(n, device=dev)
    v = torch.zeros(n, device=dev)
    p32r, p16r, mr, vr = p32.clone(), p16.clone(), m.clone(), v.clone()
    for step in (1, 2, 3):
        ops.adamw_step(p32, p16, g, m, v, lr=1e-3, step=step)
        reference.adamw_step(p32r, p16r, g, mr, vr, lr=1e-3, beta1=0.9,
                             beta2=0.95, eps=1e-8, weight_decay=0.1, step=step)
    assert rel_err(p32, p32r) < 1e-4
    assert rel_err(m, mr) < 1e-4
    assert rel_err(v, vr) < 1e-4
    assert rel_err(p16, p16r) < 1e-2


def test_fused_attention_qkv(dev):
    """Fused rope+attention off the packed qkv buffer vs the unfused ops."""
    ops = _hip()
    torch.manual_seed(3)
    B, S, Hq, Hkv, D = 2, 256, 8, 4, 128
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D, device=dev,
                      dtype=torch.bfloat16, requires_grad=True)
    cos, sin = ops.rope_tables(S, D, device=dev)
    out = ops.fused_attention_qkv(qkv, cos, sin, Hq, Hkv)
    g = torch.randn_like(out)
    out.backward(g)
    dqkv = qkv.grad.clone()

    qkv2 = qkv.detach().clone().requires_grad_(True)
    q, k, v = qkv2.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    qr = ops.rope(q.reshape(B, S, Hq, D), cos, sin)
    kr = ops.rope(k.reshape(B, S, Hkv, D), cos, sin)
    ref = ops.flash_attention(qr, kr, v.reshape(B, S, Hkv, D).contiguous(),
                              causal=True)
    ref.backward(g)
    assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(dqkv.float(), qkv2.grad.float(),
                          atol=5e-2, rtol=5e-2)


def test_swiglu_packed(dev):
    ops = _hip()
    torch.manual_seed(4)
    gu = torch.randn(64, 512, device=dev, dtype=torch.bfloat16,
                     requires_grad=True)
    out = ops.swiglu_packed(gu)
    g = torch.randn_like(out)
    out.backward(g)
    dgu = gu.grad.clone()

    gu2 = gu.detach().clone().requires_grad_(True)
    a, b = gu2.chunk(2, dim=-1)
    ref = ops.swiglu(a.contiguous(), b.contiguous())
    ref.backward(g)
    assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(dgu.float(), gu2.grad.float(), atol=2e-2, rtol=2e-2)


def test_attention_fwd_long_seq_8wave(dev):
    """S >= 8192 dispatches the 8-wave forward kernel — numerics vs the
    fp32 reference at that length."""
    ops = _hip()
    torch.manual_seed(5)
    B, S, Hq, Hkv, D = 1, 8192, 2, 1, 128
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    out = ops.flash_attention(q, k, v, causal=True)
    from torchx_amd.ops import reference

    ref = reference.attention(q.float(), k.float(), v.float(), causal=True,
                              scale=D ** -0.5)
    assert rel_err(out, ref) < 3e-2


def test_fp8_linear_numerics(dev):
    """Fp8Linear fwd/bwd vs bf16 nn.Linear within fp8 quantization error."""
    import torch.nn as nn

    from torchx_amd.parallel.fp8 import Fp8Linear

    torch.manual_seed(9)
    M, K, N = 256, 2048, 4096
    lin = nn.Linear(K, N, bias=False, dtype=torch.bfloat16).to(dev)
    f8 = Fp8Linear(K, N, weight=lin.weight)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)

    # delayed scaling warm-up: the first calls calibrate x/g amax
    for _ in range(2):
        f8(x.detach().clone().requires_grad_(True)).sum().backward()
    lin.weight.grad = None

    y8 = f8(x)
    y = lin(x2)
    assert rel_err(y8, y.float()) < 8e-2

    g = torch.randn_like(y)
    y8.backward(g)
    y.backward(g)
    # both dgrad operands are e4m3-quantized (delayed scales): ~0.2-0.3
    # worst-case relative error on random data
    assert torch.isfinite(x.grad.float()).all()
    assert rel_err(x.grad, x2.grad.float()) < 4e-1
    # weight grad accumulated on the SHARED parameter by both backwards;
    # compare halves via fresh run instead
    lin.weight.grad = None
    x3 = x.detach().clone().requires_grad_(True)
    f8(x3).backward(g)
    dw8 = lin.weight.grad.clone()
    lin.weight.grad = None
    x4 = x.detach().clone().requires_grad_(True)
    lin(x4).backward(g)
    assert torch.isfinite(dw8.float()).all()
    assert rel_err(dw8, lin.weight.grad.float()) < 4e-1


def test_fp8_model_step(dev):
    """Full llama gpu_tiny step with fp8 linears converges."""
    from torchx_amd.models.llama import LlamaModel, llama_gpu_tiny
    from torchx_amd.parallel import (
        FlatAdamW, FlatDDP, FlatParams, convert_to_fp8,
    )

    cfg = llama_gpu_tiny()
    model = LlamaModel(cfg, device=dev)
    convert_to_fp8(model)
    flat = FlatParams(model, dev)
    ddp = FlatDDP(flat)
    opt = FlatAdamW(flat, lr=1e-3)
    torch.manual_seed(7)
    tokens = torch.randint(0, cfg.vocab_size, (2, 128), device=dev)
    targets = torch.roll(tokens, -1, 1)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        loss = model(tokens, targets)
        loss.backward()
        ddp.finish()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] - 0.5, losses

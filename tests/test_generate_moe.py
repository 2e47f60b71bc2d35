"""Mixtral serving path: cached MoE decode must agree with a full
re-forward (CPU reference kernels here, HIP kernels in the gpu test)."""

import pytest
import torch

from torchx_amd.models.generate_moe import (
    KVCache, decode_step_moe, generate_moe, prefill_moe,
)
from torchx_amd.models.mixtral import MixtralModel, mixtral_tiny


def test_cached_moe_decode_matches_full_forward():
    torch.manual_seed(0)
    cfg = mixtral_tiny()
    model = MixtralModel(cfg)
    B, S0 = 2, 12
    tokens = torch.randint(0, cfg.vocab_size, (B, S0))
    caches = [KVCache.empty(cfg, B, S0 + 6, torch.device("cpu"))
              for _ in range(cfg.num_layers)]
    pre = prefill_moe(model, tokens, caches)
    with torch.no_grad():
        full = model(tokens)[:, -1]
    assert torch.allclose(pre.float(), full.float(), atol=3e-2), (
        (pre - full).abs().max())

    cur = tokens
    nxt = pre.argmax(-1, keepdim=True)
    for _ in range(3):
        dec = decode_step_moe(model, nxt, caches)
        cur = torch.cat([cur, nxt], dim=1)
        with torch.no_grad():
            full = model(cur)[:, -1]
        assert torch.allclose(dec.float(), full.float(), atol=4e-2), (
            (dec - full).abs().max())
        nxt = dec.argmax(-1, keepdim=True)


def test_generate_moe_shapes():
    torch.manual_seed(1)
    cfg = mixtral_tiny()
    model = MixtralModel(cfg)
    tokens = torch.randint(0, cfg.vocab_size, (2, 8))
    out = generate_moe(model, tokens, max_new_tokens=4)
    assert out.shape == (2, 12)
    assert torch.equal(out[:, :8], tokens)
    out2 = generate_moe(model, tokens, max_new_tokens=4)
    assert torch.equal(out, out2)  # greedy determinism
    out3 = generate_moe(model, tokens, max_new_tokens=3, temperature=0.7,
                        top_k=10)
    assert out3.shape == (2, 11)


@pytest.mark.gpu
def test_generate_moe_gpu_end_to_end():
    from torchx_amd.models.mixtral import mixtral_gpu_tiny

    dev = torch.device("cuda:0")
    torch.manual_seed(2)
    cfg = mixtral_gpu_tiny()
    model = MixtralModel(cfg, device=dev)
    tokens = torch.randint(0, cfg.vocab_size, (2, 32), device=dev)
    out = generate_moe(model, tokens, max_new_tokens=6)
    assert out.shape == (2, 38)
    # cached decode logits agree with a full re-forward on the HIP path
    caches = [KVCache.empty(cfg, 2, 48, dev) for _ in range(cfg.num_layers)]
    prefill_moe(model, tokens, caches)
    nxt = tokens[:, -1:]
    dec = decode_step_moe(model, nxt, caches)
    with torch.no_grad():
        full = model(torch.cat([tokens, nxt], 1))[:, -1]
    err = (dec.float() - full.float()).abs().max().item()
    scale = full.float().abs().max().item() + 1e-6
    assert err < 5e-2 * scale, (err, scale)


def test_generate_app_moe_cpu(capsys):
    from torchx_amd.apps import generate_main

    rc = generate_main.main(["--model", "mixtral_tiny", "--batch", "2",
                             "--prompt-len", "10", "--new-tokens", "3"])
    assert rc == 0
    import json as _json

    rec = _json.loads(capsys.readouterr().out.strip().splitlines()[-1])
    assert rec["model"] == "mixtral_tiny" and rec["tokens_per_second"] > 0
    # --graph is refused for MoE models
    rc = generate_main.main(["--model", "mixtral_tiny", "--graph",
                             "--prompt-len", "8", "--new-tokens", "2"])
    assert rc == 1


def test_continuous_batcher_moe_matches_per_sequence():
    """Ragged MoE decode: two Mixtral prompts of different lengths in one
    batcher produce exactly their solo greedy streams."""
    from torchx_amd.models.generate import ContinuousBatcher

    torch.manual_seed(14)
    cfg = mixtral_tiny()
    model = MixtralModel(cfg)
    p1 = torch.randint(0, cfg.vocab_size, (9,))
    p2 = torch.randint(0, cfg.vocab_size, (14,))
    solo1 = generate_moe(model, p1.reshape(1, -1), 4)[0, 9:].tolist()
    solo2 = generate_moe(model, p2.reshape(1, -1), 4)[0, 14:].tolist()
    cb = ContinuousBatcher(model, max_batch=2, max_len=32,
                           prefill_fn=prefill_moe,
                           decode_fn=decode_step_moe)
    got1 = [int(cb.admit(0, p1))]
    got2 = [int(cb.admit(1, p2))]
    for _ in range(3):
        toks = cb.step()
        got1.append(int(toks[0]))
        got2.append(int(toks[1]))
    assert got1 == solo1, (got1, solo1)
    assert got2 == solo2, (got2, solo2)

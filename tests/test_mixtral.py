"""Mixtral MoE tests: local routing math vs a naive reference, training
step, and a 2-process gloo expert-parallel numerics check (EP all-to-all
vs the single-process result)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

from torchx_amd.models.mixtral import (
    Expert,
    MixtralConfig,
    MixtralModel,
    MoELayer,
    mixtral_tiny,
)

REPO = Path(__file__).resolve().parent.parent


def test_moe_matches_naive_reference():
    torch.manual_seed(0)
    cfg = mixtral_tiny()
    moe = MoELayer(cfg)
    x = torch.randn(2, 16, cfg.hidden_size, dtype=torch.bfloat16)
    out = moe(x)

    # naive: for each token run its top-k experts directly
    xt = x.reshape(-1, cfg.hidden_size)
    logits = moe.router(xt).float()
    w, e = torch.topk(logits, cfg.top_k, dim=-1)
    w = torch.softmax(w, dim=-1).to(x.dtype)
    ref = torch.zeros_like(xt)
    for t in range(xt.shape[0]):
        for j in range(cfg.top_k):
            ref[t] += w[t, j] * moe.local_experts[int(e[t, j])](xt[t:t + 1])[0]
    ref = ref.reshape(x.shape)
    err = (out.float() - ref.float()).abs().max().item()
    assert err < 5e-2, f"moe mismatch {err}"


def test_mixtral_train_step_decreases_loss():
    torch.manual_seed(0)
    cfg = mixtral_tiny()
    from torchx_amd.parallel import FlatAdamW, FlatParams

    model = MixtralModel(cfg)
    flat = FlatParams(model, torch.device("cpu"))
    opt = FlatAdamW(flat, lr=1e-3)
    tokens = torch.randint(0, cfg.vocab_size, (2, 32))
    targets = torch.randint(0, cfg.vocab_size, (2, 32))
    losses = []
    for _ in range(4):
        opt.zero_grad()
        loss = model(tokens, targets)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses


EP_WORKER = r"""
import os, sys
sys.path.insert(0, %(repo)r)
import torch
import torch.distributed as dist
from torchx_amd.models.mixtral import MoELayer, mixtral_tiny

dist.init_process_group("gloo")
rank = dist.get_rank()
ws = dist.get_world_size()
torch.manual_seed(7)
cfg = mixtral_tiny()  # 4 experts

# full (non-EP) layer with all experts, same seed on both ranks
full = MoELayer(cfg, ep_size=1)

# EP layer: rank owns cfg.num_experts//ws experts, weights copied from full
ep = MoELayer(cfg, ep_group=None, ep_size=ws, ep_rank=rank)
ep.router.load_state_dict(full.router.state_dict())
epr = cfg.num_experts // ws
for i in range(epr):
    ep.local_experts[i].load_state_dict(
        full.local_experts[rank * epr + i].state_dict())

torch.manual_seed(11)  # same input everywhere
x = torch.randn(1, 16, cfg.hidden_size, dtype=torch.bfloat16)
out_full = full(x)
out_ep = ep(x)
err = (out_full.float() - out_ep.float()).abs().max().item()
assert err < 5e-2, f"rank {rank}: EP mismatch {err}"
if rank == 0:
    print("EP_NUMERICS_OK", flush=True)
dist.destroy_process_group()
"""


def test_moe_expert_parallel_gloo(tmp_path):
    script = tmp_path / "ep_worker.py"
    script.write_text(EP_WORKER % {"repo": str(REPO)})
    proc = subprocess.run(
        [
            sys.executable, "-m", "torchx_amd.agent",
            "--nnodes", "1", "--nproc-per-node", "2",
            "--rdzv-endpoint", "127.0.0.1:0",
            "--rdzv-id", "test_ep", "--max-restarts", "0",
            str(script),
        ],
        cwd=str(REPO),
        env={**os.environ, "MASTER_ADDR": "127.0.0.1"},
        capture_output=True,
        timeout=300,
    )
    out = proc.stdout.decode() + proc.stderr.decode()
    assert proc.returncode == 0, out[-4000:]
    assert "EP_NUMERICS_OK" in out, out[-4000:]

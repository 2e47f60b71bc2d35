"""Trainer app tests: checkpoint/resume over fsspec and tracker artifact
wiring (parity: the reference's lightning example conventions)."""

import os

import pytest
import torch

from torchx_amd.apps import trainer


def test_train_checkpoints_and_resumes(tmp_path, monkeypatch):
    ckpt = str(tmp_path / "ckpts")
    rc = trainer.main([
        "--steps", "4", "--checkpoint-every", "2",
        "--checkpoint-dir", ckpt, "--seq-len", "32", "--micro-batch", "1",
    ])
    assert rc == 0
    files = sorted(os.listdir(ckpt))
    assert "step_2.pt" in files and "step_4.pt" in files

    # resume continues from latest step: running with --steps 4 again is a
    # no-op (start == steps), with --steps 6 runs two more steps
    rc = trainer.main([
        "--steps", "6", "--resume", "--checkpoint-every", "100",
        "--checkpoint-dir", ckpt, "--seq-len", "32", "--micro-batch", "1",
    ])
    assert rc == 0
    assert "step_6.pt" in os.listdir(ckpt)


def test_resume_restores_state(tmp_path):
    ckpt = str(tmp_path / "c")
    torch.manual_seed(0)
    trainer.main(["--steps", "3", "--checkpoint-every", "3",
                  "--checkpoint-dir", ckpt, "--seq-len", "32",
                  "--micro-batch", "1"])
    latest = trainer._latest_checkpoint(ckpt)
    assert latest and latest.endswith("step_3.pt")

    from torchx_amd.models.llama import LlamaModel, llama_tiny
    from torchx_amd.parallel import FlatAdamW, FlatParams

    cfg = llama_tiny()
    dev = torch.device("cpu")
    model = LlamaModel(cfg, device=dev)
    flat = FlatParams(model, dev)
    opt = FlatAdamW(flat)
    step = trainer.load_checkpoint(latest, model, opt)
    assert step == 3
    assert opt.step_count == 3
    # flat buffer views must reflect the loaded weights
    w = dict(model.named_parameters())["embed.weight"]
    assert torch.equal(
        w.data.reshape(-1),
        flat.flat_p16["decay"][: w.numel()],
    )


def test_tracker_artifact_written(tmp_path, monkeypatch):
    tracker_root = tmp_path / "tracker"
    cfg_file = tmp_path / "tracker.cfg"
    cfg_file.write_text(f"[fsspec]\nroot_path = {tracker_root}\n")
    monkeypatch.setenv("TORCHX_JOB_ID", "local_cwd://test/app_123")
    monkeypatch.setenv("TORCHX_TRACKERS",
                       "torchx_amd.tracker.fsspec:create")
    monkeypatch.setenv(
        "TORCHX_TRACKER_TORCHX_AMD_TRACKER_FSSPEC:CREATE_CONFIG".replace(
            ":", "_").replace(".", "_").upper(), str(cfg_file))
    ckpt = str(tmp_path / "ck")
    rc = trainer.main(["--steps", "1", "--checkpoint-every", "1",
                       "--checkpoint-dir", ckpt, "--seq-len", "32",
                       "--micro-batch", "1"])
    assert rc == 0


def test_datapreproc_synthetic(tmp_path):
    import numpy as np

    from torchx_amd.apps import datapreproc

    out = tmp_path / "tokens.bin"
    rc = datapreproc.main(["--synthetic", "20", "--output", str(out),
                           "--vocab-size", "512"])
    assert rc == 0
    ids = np.frombuffer(out.read_bytes(), dtype=np.uint32)
    assert len(ids) > 20
    assert ids.max() < 512
    assert (ids == 1).sum() == 20  # one EOS per doc

"""The classifier example app family (reference parity: the Lightning
example suite, torchx/examples/apps/lightning/) — runs end-to-end on CPU:
data creation, training, checkpoint resume, export; plus a 2-process
gloo DDP run through torch.distributed.run."""

import os
import subprocess

import pytest
import sys
from pathlib import Path

import torch

pytestmark = pytest.mark.subprocess_heavy

REPO = Path(__file__).resolve().parent.parent


def test_train_end_to_end(tmp_path):
    from torchx_amd.examples.apps.classifier.train import main

    out = tmp_path / "out"
    rc = main(["--epochs", "1", "--num_samples", "64",
               "--batch_size", "16", "--output_path", str(out)])
    assert rc == 0
    assert (out / "ckpt_epoch0.pt").exists()
    assert (out / "model_scripted.pt").exists()
    # exported model is loadable and runs
    m = torch.jit.load(str(out / "model_scripted.pt"))
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 8)


def test_train_resume(tmp_path):
    from torchx_amd.examples.apps.classifier.train import main

    out = tmp_path / "out"
    assert main(["--epochs", "1", "--num_samples", "32",
                 "--output_path", str(out), "--skip_export"]) == 0
    ck = out / "ckpt_epoch0.pt"
    assert ck.exists()
    assert main(["--epochs", "2", "--num_samples", "32",
                 "--output_path", str(out), "--skip_export",
                 "--load_path", str(ck)]) == 0
    assert (out / "ckpt_epoch1.pt").exists()


def test_data_module_roundtrip(tmp_path):
    from torchx_amd.examples.apps.classifier.data import (
        ImageDataModule, SyntheticImages, create_random_data, load_data,
    )

    path = create_random_data(str(tmp_path / "d"), n=48, num_classes=4)
    ds, nc = load_data(path)
    assert nc == 4
    assert len(ds) == 48
    dm = ImageDataModule(ds, batch_size=8)
    xb, yb = next(iter(dm.train_loader()))
    assert xb.shape == (8, 3, 64, 64)
    assert yb.dtype == torch.int64


def test_train_ddp_gloo(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes", "1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--standalone", "-m", "torchx_amd.examples.apps.classifier.train",
         "--epochs", "1", "--num_samples", "48", "--batch_size", "8",
         "--output_path", str(tmp_path / "out"), "--skip_export"],
        env=env, capture_output=True, text=True, timeout=300, cwd=str(REPO),
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert (tmp_path / "out" / "ckpt_epoch0.pt").exists()


def test_interpret_app(tmp_path):
    """interpret.py (reference lightning/interpret.py counterpart): train a
    checkpoint, attribute a batch, artifacts land on the output path."""
    import json

    import torch

    from torchx_amd.examples.apps.classifier import interpret, train

    from torchx_amd.examples.apps.classifier.data import create_random_data

    data = create_random_data(str(tmp_path / "data"), n=32)
    out = str(tmp_path / "out")
    rc = train.main(["--dataset_path", data, "--output_path", out,
                     "--epochs", "1", "--batch_size", "8",
                     "--skip_export"])
    assert rc == 0
    ckpts = sorted((tmp_path / "out").glob("ckpt_epoch*.pt"))
    assert ckpts
    att_out = str(tmp_path / "attr")
    rc = interpret.main(["--load-path", str(ckpts[-1]),
                         "--data-path", data,
                         "--output-path", att_out, "--batch", "4"])
    assert rc == 0
    blob = torch.load(tmp_path / "attr" / "attributions.pt",
                      weights_only=True)
    assert blob["saliency"].shape[0] == 4
    assert blob["occlusion"].shape[0] == 4
    summary = json.loads((tmp_path / "attr" / "summary.json").read_text())
    assert summary["n"] == 4

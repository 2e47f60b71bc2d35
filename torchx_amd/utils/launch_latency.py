"""Launch -> first-step latency of the dist.ddp path (half of the
BASELINE.json headline metric): time from Runner.run_component() to the
first training-step line appearing in the app's logs.

The payload is the bundled trainer on the tiny (CPU/gloo) or gpu_tiny
(RCCL) config; the measured span covers component materialization, dryrun,
Popen fan-out, agent start, c10d rendezvous, process-group init and the
first fwd+bwd+step — everything the reference delegates to torchrun
(torchx/components/dist.py:262).

Importable (bench.py embeds the measurement in its JSON line so the
driver gets a record of BOTH halves of the metric) and runnable via
tools/launch_latency.py.
"""

from __future__ import annotations

import sys
import tempfile
import time
from typing import Optional


def measure_launch_latency(
    nproc: int = 2,
    model: Optional[str] = None,
    timeout: float = 300.0,
    log_dir: Optional[str] = None,
) -> dict:
    import torch

    from torchx_amd.runner import get_runner
    from torchx_amd.specs import AppState, is_terminal

    model = model or ("gpu_tiny" if torch.cuda.is_available() else "tiny")
    log_dir = log_dir or tempfile.mkdtemp(prefix="launch-lat-")

    first_step = None
    failed = None
    with get_runner("latency") as runner:
        t0 = time.perf_counter()
        handle = runner.run_component(
            "dist.ddp",
            ["-j", f"1x{nproc}", "-m", "torchx_amd.apps.trainer",
             "--steps", "1", "--model", model,
             "--seq-len", "64", "--micro-batch", "1"],
            scheduler="local_cwd",
            cfg={"log_dir": log_dir, "auto_set_hip_visible_devices": False},
        )
        t_submitted = time.perf_counter() - t0

        role = None
        deadline = time.time() + timeout
        while time.time() < deadline:
            status = runner.status(handle)
            if role is None and status and status.roles:
                role = status.roles[0].role
            lines = []
            try:
                if role:
                    lines = list(runner.log_lines(handle, role, 0))
            except Exception:  # noqa: BLE001 — logs not there yet
                pass
            if any("step 1 loss" in ln for ln in lines):
                first_step = time.perf_counter() - t0
                break
            if status and is_terminal(status.state):
                if status.state != AppState.SUCCEEDED:
                    failed = str(status)
                    break
                first_step = time.perf_counter() - t0
                break
            time.sleep(0.05)
        runner.wait(handle, wait_interval=0.5)

    if failed:
        print(f"launch-latency app failed: {failed}", file=sys.stderr)

    return {
        "metric": "launch_to_first_step_seconds",
        "value": first_step,
        "submit_seconds": t_submitted,
        "nproc": nproc,
        "model": model,
        "backend": "nccl" if torch.cuda.is_available() else "gloo",
        "higher_is_better": False,
    }

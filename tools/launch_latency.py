#!/usr/bin/env python3
"""Measure launch -> first-step latency of the dist.ddp path (the BASELINE
headline latency metric): time from Runner.run_component() to the first
training-step line appearing in the app's logs.

The payload is the bundled trainer on the tiny (CPU/gloo) or gpu_tiny
(RCCL) config; the measured span covers component materialization, dryrun,
Popen fan-out, agent start, c10d rendezvous, process-group init and the
first fwd+bwd+step — everything the reference delegates to torchrun.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--nproc", type=int, default=2)
    p.add_argument("--model", type=str, default=None,
                   help="default: tiny on CPU, gpu_tiny when a GPU is there")
    p.add_argument("--timeout", type=float, default=300.0)
    p.add_argument("--log-dir", type=str, default=None)
    args = p.parse_args()

    import torch

    from torchx_amd.runner import get_runner
    from torchx_amd.specs import AppState, is_terminal

    model = args.model or (
        "gpu_tiny" if torch.cuda.is_available() else "tiny"
    )
    import tempfile

    log_dir = args.log_dir or tempfile.mkdtemp(prefix="launch-lat-")

    with get_runner("latency") as runner:
        t0 = time.perf_counter()
        handle = runner.run_component(
            "dist.ddp",
            ["-j", f"1x{args.nproc}", "-m", "torchx_amd.apps.trainer",
             "--steps", "1", "--model", model,
             "--seq-len", "64", "--micro-batch", "1"],
            scheduler="local_cwd",
            cfg={"log_dir": log_dir, "auto_set_hip_visible_devices": False},
        )
        t_submitted = time.perf_counter() - t0

        first_step = None
        role = None
        deadline = time.time() + args.timeout
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
                    print(f"app failed: {status}", file=sys.stderr)
                    return 1
                first_step = time.perf_counter() - t0
                break
            time.sleep(0.05)
        runner.wait(handle, wait_interval=0.5)

    result = {
        "metric": "launch_to_first_step_seconds",
        "value": first_step,
        "submit_seconds": t_submitted,
        "nproc": args.nproc,
        "model": model,
        "backend": "nccl" if torch.cuda.is_available() else "gloo",
        "higher_is_better": False,
    }
    print(json.dumps(result), flush=True)
    return 0 if first_step is not None else 1


if __name__ == "__main__":
    sys.exit(main())

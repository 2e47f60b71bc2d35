#!/usr/bin/env python3
"""CLI wrapper for torchx_amd.utils.launch_latency (the BASELINE headline
latency metric). See that module for what the measured span covers."""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--nproc", type=int, default=2)
    p.add_argument("--model", type=str, default=None,
                   help="default: tiny on CPU, gpu_tiny when a GPU is there")
    p.add_argument("--timeout", type=float, default=300.0)
    p.add_argument("--log-dir", type=str, default=None)
    args = p.parse_args()

    from torchx_amd.utils.launch_latency import measure_launch_latency

    result = measure_launch_latency(
        nproc=args.nproc, model=args.model, timeout=args.timeout,
        log_dir=args.log_dir,
    )
    print(json.dumps(result), flush=True)
    return 0 if result["value"] is not None else 1


if __name__ == "__main__":
    sys.exit(main())

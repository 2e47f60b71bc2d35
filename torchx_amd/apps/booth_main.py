"""Booth-function app for HPO smoke tests (parity:
torchx/apps/utils/booth_main.py): writes f(x1,x2) via the result tracker."""

import argparse

from torchx_amd.runtime.tracking import FsspecResultTracker


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--x1", type=float, required=True)
    p.add_argument("--x2", type=float, required=True)
    p.add_argument("--trial_idx", type=int, default=0)
    p.add_argument("--tracker_base", type=str, required=True)
    args = p.parse_args()
    val = (args.x1 + 2 * args.x2 - 7) ** 2 + (2 * args.x1 + args.x2 - 5) ** 2
    tracker = FsspecResultTracker(args.tracker_base)
    tracker[args.trial_idx] = {"booth_eval": val}


if __name__ == "__main__":
    main()

from .api import AppRun, TrackerBase, app_run_from_env, trackers_from_environ  # noqa: F401


def tracker_cli(args) -> int:
    """`torchx tracker list jobs|metadata|artifacts` (parity:
    torchx/cli/cmd_tracker.py)."""
    from torchx_amd.runner.config import get_configured_trackers

    configured = get_configured_trackers()
    if not configured:
        print("no trackers configured ([torchx:tracker] in .torchxconfig)")
        return 1
    from .fsspec import FsspecTracker, create as create_fsspec

    name, cfg = next(iter(configured.items()))
    tracker = create_fsspec(cfg) if cfg else None
    if tracker is None:
        print(f"tracker {name} has no config")
        return 1
    if args.entity == "jobs":
        for run_id in tracker.run_ids():
            print(run_id)
    elif args.entity == "metadata":
        for k, v in tracker.metadata(args.run_id).items():
            print(f"{k}\t{v}")
    elif args.entity == "artifacts":
        for name_, path in tracker.artifacts(args.run_id).items():
            print(f"{name_}\t{path}")
    return 0

"""torchx CLI (parity: torchx/cli/main.py — subcommands run, builtins,
cancel, configure, delete, describe, list, log, runopts, status, tracker)."""

from __future__ import annotations

import argparse
import inspect
import json
import logging
import os
import sys
import threading
import time
from typing import Dict, List, Optional

from torchx_amd.runner import Runner, get_runner
from torchx_amd.runner import config as torchx_config
from torchx_amd.schedulers import (
    get_default_scheduler_name,
    get_scheduler_factories,
)
from torchx_amd.specs import AppState, parse_app_handle
from torchx_amd.specs.finder import (
    get_builtin_source,
    get_components,
)

logger = logging.getLogger(__name__)

_COLORS = ["\033[32m", "\033[33m", "\033[34m", "\033[35m", "\033[36m"]
_RESET = "\033[0m"


def _parse_run_args(args: List[str]) -> (str, List[str]):
    """Split ``component_name [component args...]``; default component from
    .torchxconfig [cli:run] component= (parity: cmd_run.py:119)."""
    if args and not args[0].startswith("-"):
        return args[0], args[1:]
    default = torchx_config.get_config("cli", "run", "component")
    if default:
        return default, args
    raise SystemExit(
        "error: no component name given and no default configured "
        "([cli:run] component= in .torchxconfig)"
    )


def _component_defaults() -> Dict[str, Dict[str, str]]:
    return torchx_config.load_sections("component")


# ---------------------------------------------------------------------------
# subcommands
# ---------------------------------------------------------------------------


def cmd_run(args: argparse.Namespace) -> int:
    # context name for downstream tracking (parity: cli/cmd_run.py:446)
    os.environ.setdefault("TORCHX_CONTEXT_NAME", "cli_run")
    stdin_scheduler_args: Optional[Dict[str, object]] = None
    if getattr(args, "stdin", False):
        # JSON run spec from stdin: {"component": ..., "component_args":
        # [...], "scheduler": ..., "scheduler_args": {...}, "dryrun": bool}
        # (reference parity: cli/cmd_run.py:366-399)
        try:
            spec = json.load(sys.stdin)
        except (json.JSONDecodeError, EOFError):
            print("invalid JSON on stdin for `torchx run`", file=sys.stderr)
            return 1
        if not isinstance(spec, dict) or "component" not in spec:
            print("stdin JSON must be a dict with a `component` key",
                  file=sys.stderr)
            return 1
        args.component_name_and_args = (
            [spec["component"]] + [str(a) for a in
                                   spec.get("component_args", [])]
        )
        args.scheduler = spec.get("scheduler", args.scheduler)
        # keep the JSON dict as-is — flattening to "k=v,k=v" would corrupt
        # values containing ',' or '=' (mounts, dict-typed opts)
        stdin_scheduler_args = dict(spec.get("scheduler_args", {}))
        args.dryrun = bool(spec.get("dryrun", args.dryrun))
    component, comp_args = _parse_run_args(args.component_name_and_args)
    # precedence: -s flag > [cli:run] scheduler= > built-in default
    # (reference parity: cli/argparse_util.py torchxconfig-backed defaults)
    scheduler = (args.scheduler
                 or torchx_config.get_config("cli", "run", "scheduler")
                 or "local_cwd")
    runner = get_runner(component_defaults=_component_defaults())
    opts = runner.scheduler_run_opts(scheduler)
    cfg = opts.cfg_from_str(args.scheduler_args)
    if stdin_scheduler_args:
        for k, v in stdin_scheduler_args.items():
            opt = opts.get(k)
            cfg[k] = opt.cast(v) if opt is not None else v
    if args.dryrun:
        info = runner.dryrun_component(
            component, comp_args, scheduler, cfg=cfg,
            workspace=args.workspace,
        )
        print("=== APPLICATION ===")
        print(info._app)
        print("=== SCHEDULER REQUEST ===")
        print(info)
        return 0
    handle = runner.run_component(
        component, comp_args, scheduler, cfg=cfg, workspace=args.workspace,
    )
    print(handle, flush=True)
    if not args.wait and scheduler.startswith("local"):
        args.wait = True  # local runs attach by default (reference :321)
    if args.wait:
        return _wait_and_exit(runner, handle, log=args.log or
                              scheduler.startswith("local"))
    return 0


def _wait_and_exit(runner: Runner, handle: str, log: bool = False) -> int:
    log_thread = None
    if log:
        log_thread = threading.Thread(
            target=_stream_logs, args=(runner, handle), daemon=True
        )
        log_thread.start()
    status = runner.wait(handle, wait_interval=1.0)
    if log_thread:
        log_thread.join(timeout=10)
    if status is None:
        print("app not found", file=sys.stderr)
        return 1
    print(status.format())
    return 0 if status.state == AppState.SUCCEEDED else 1


def _stream_logs(runner: Runner, handle: str) -> None:
    app = None
    deadline = time.time() + 30
    while app is None and time.time() < deadline:
        app = runner.describe(handle)
        if app is None:
            time.sleep(0.5)
    # one thread per (role, replica) with colored prefixes (cmd_log.py:98)
    desc = runner.status(handle)
    threads = []
    try:
        scheduler, _, app_id = parse_app_handle(handle)
        roles = {rs.role: len(rs.replicas) for rs in (desc.roles if desc else [])}
        if not roles:
            return
        i = 0
        for role, n in roles.items():
            for k in range(n):
                color = _COLORS[i % len(_COLORS)] if sys.stdout.isatty() else ""
                reset = _RESET if color else ""
                prefix = f"{color}{role}/{k}{reset} "

                def pump(role=role, k=k, prefix=prefix):
                    try:
                        for line in runner.log_lines(
                            handle, role, k, should_tail=True
                        ):
                            print(prefix + line, flush=True)
                    except Exception:  # noqa: BLE001
                        pass

                t = threading.Thread(target=pump, daemon=True)
                t.start()
                threads.append(t)
                i += 1
        for t in threads:
            t.join()
    except Exception:  # noqa: BLE001
        pass


def cmd_log(args: argparse.Namespace) -> int:
    # identifier: SCHEDULER://SESSION/APP_ID[/ROLE[/REPLICA]]
    ident = args.identifier
    parts = ident.split("/")
    runner = get_runner()
    role: Optional[str] = None
    replica: Optional[int] = None
    # scheduler://session/app_id[/role[/replica]]
    base = "/".join(parts[:4]) if len(parts) > 4 else ident
    extra = parts[4:] if len(parts) > 4 else []
    if extra:
        role = extra[0]
        if len(extra) > 1:
            replica = int(extra[1])
    status = runner.status(base)
    if status is None:
        print(f"app not found: {base}", file=sys.stderr)
        return 1
    roles = {rs.role: len(rs.replicas) for rs in status.roles}
    targets = []
    for r, n in roles.items():
        if role and r != role:
            continue
        for k in range(n):
            if replica is not None and k != replica:
                continue
            targets.append((r, k))
    from torchx_amd.utils.log_tee import print_log_lines

    # one thread per (role, replica), colored interleaved prefixes
    # (reference parity: cli/cmd_log.py:144-163)
    print_log_lines(
        targets,
        lambda r, k: runner.log_lines(base, r, k, regex=args.regex,
                                      should_tail=args.follow),
    )
    return 0


def cmd_status(args: argparse.Namespace) -> int:
    status = get_runner().status(args.app_handle)
    if status is None:
        print(f"app not found: {args.app_handle}", file=sys.stderr)
        return 1
    print(status.format())
    return 0


def cmd_describe(args: argparse.Namespace) -> int:
    app = get_runner().describe(args.app_handle)
    if app is None:
        print(f"app not found: {args.app_handle}", file=sys.stderr)
        return 1
    print(app)
    return 0


def cmd_cancel(args: argparse.Namespace) -> int:
    get_runner().cancel(args.app_handle)
    print(f"cancelled {args.app_handle}")
    return 0


def cmd_delete(args: argparse.Namespace) -> int:
    get_runner().delete(args.app_handle)
    print(f"deleted {args.app_handle}")
    return 0


def cmd_list(args: argparse.Namespace) -> int:
    responses = get_runner().list(args.scheduler)
    fmt = "{:<40} {:<12}"
    print(fmt.format("APP HANDLE", "STATUS"))
    for r in responses:
        print(fmt.format(r.app_handle, str(r.state)))
    return 0


def cmd_runopts(args: argparse.Namespace) -> int:
    runner = get_runner()
    scheds = [args.scheduler] if args.scheduler else runner.scheduler_backends()
    for s in scheds:
        print(f"{s}:")
        try:
            print(runner.scheduler_run_opts(s))
        except Exception as e:  # noqa: BLE001
            print(f"  (unavailable: {e})")
    return 0


def cmd_builtins(args: argparse.Namespace) -> int:
    comps = get_components()
    if args.print:
        print(get_builtin_source(args.print))
        return 0
    print(f"Found {len(comps)} builtin components:")
    for i, (name, comp) in enumerate(sorted(comps.items()), 1):
        print(f"  {i}. {name} - {comp.description}")
    return 0


def cmd_configure(args: argparse.Namespace) -> int:
    path = os.path.join(os.getcwd(), ".torchxconfig")
    torchx_config.dump(path, schedulers=args.schedulers.split(",")
                       if args.schedulers else None)
    print(f"wrote {path}")
    return 0


def cmd_tracker(args: argparse.Namespace) -> int:
    from torchx_amd.tracker import tracker_cli

    return tracker_cli(args)


# ---------------------------------------------------------------------------
# parser
# ---------------------------------------------------------------------------


def create_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="torchx",
        description="torchx_amd: MI355X-native distributed job launcher",
    )
    parser.add_argument("--log_level", type=int, default=logging.INFO)
    parser.add_argument("--version", action="version",
                        version="torchx-amd-0.1.0")
    sub = parser.add_subparsers(title="sub-commands", dest="cmd")

    default_sched = "local_cwd"

    p = sub.add_parser("run", help="run a component as a job")
    p.add_argument("-s", "--scheduler", type=str, default=None,
                   help=f"scheduler (default: [cli:run] scheduler= from "
                        f".torchxconfig, else {default_sched})")
    p.add_argument("-cfg", "--scheduler_args", type=str, default="",
                   help="scheduler runopts, e.g. k1=v1,k2=v2")
    p.add_argument("--dryrun", action="store_true")
    p.add_argument("--stdin", action="store_true",
                   help="read a JSON run spec from stdin")
    p.add_argument("--wait", action="store_true")
    p.add_argument("--log", action="store_true")
    p.add_argument("--workspace", type=str, default=None)
    p.add_argument("component_name_and_args", nargs=argparse.REMAINDER)
    p.set_defaults(func=cmd_run)

    p = sub.add_parser("status", help="app status")
    p.add_argument("app_handle")
    p.set_defaults(func=cmd_status)

    p = sub.add_parser("describe", help="describe an app")
    p.add_argument("app_handle")
    p.set_defaults(func=cmd_describe)

    p = sub.add_parser("cancel", help="cancel an app")
    p.add_argument("app_handle")
    p.set_defaults(func=cmd_cancel)

    p = sub.add_parser("delete", help="delete an app")
    p.add_argument("app_handle")
    p.set_defaults(func=cmd_delete)

    p = sub.add_parser("list", help="list apps on a scheduler")
    p.add_argument("-s", "--scheduler", type=str, default=default_sched)
    p.set_defaults(func=cmd_list)

    p = sub.add_parser("log", help="print app logs")
    p.add_argument("--regex", type=str, default=None)
    p.add_argument("-f", "--follow", action="store_true")
    p.add_argument("identifier",
                   help="scheduler://session/app_id[/role[/replica]]")
    p.set_defaults(func=cmd_log)

    p = sub.add_parser("runopts", help="print scheduler runopts")
    p.add_argument("scheduler", nargs="?", default=None)
    p.set_defaults(func=cmd_runopts)

    p = sub.add_parser("builtins", help="list builtin components")
    p.add_argument("--print", type=str, default=None,
                   help="print the source of a builtin")
    p.set_defaults(func=cmd_builtins)

    p = sub.add_parser("configure", help="write a .torchxconfig template")
    p.add_argument("-s", "--schedulers", type=str, default=None)
    p.set_defaults(func=cmd_configure)

    p = sub.add_parser("tracker", help="experiment tracker queries")
    p.add_argument("action", choices=["list"], default="list", nargs="?")
    p.add_argument("entity", choices=["jobs", "metadata", "artifacts"],
                   nargs="?", default="jobs")
    p.add_argument("--run_id", type=str, default=None)
    p.set_defaults(func=cmd_tracker)

    _add_entry_point_cmds(sub)

    return parser


def _add_entry_point_cmds(sub: argparse._SubParsersAction) -> None:
    """Custom/override subcommands from entry points (reference parity:
    torchx/cli/main.py:64 group ``torchx.cli.cmds``). Each entry point
    loads to a SubCommand-style object with ``add_arguments(parser)`` and
    ``run(args)``; an entry point named like a builtin replaces it."""
    try:
        from importlib.metadata import entry_points

        eps = entry_points()
    except Exception:  # noqa: BLE001
        return
    for group in ("torchx_amd.cli.cmds", "torchx.cli.cmds"):
        found = (eps.select(group=group) if hasattr(eps, "select")
                 else eps.get(group, []))
        for ep in found:
            try:
                obj = ep.load()
                cmd = obj() if isinstance(obj, type) else obj
                # override: drop a builtin parser of the same name
                sub._name_parser_map.pop(ep.name, None)
                p = sub.add_parser(
                    ep.name,
                    help=(inspect.getdoc(cmd) or "").split("\n")[0] or None,
                )
                add_args = getattr(cmd, "add_arguments", None)
                if add_args:
                    add_args(p)
                p.set_defaults(func=cmd.run)
            except Exception as e:  # noqa: BLE001 — a broken CLI plugin
                # must not take down the whole CLI
                logger.warning("skipping CLI plugin %s:%s: %s",
                               group, ep.name, e)


def main(argv: Optional[List[str]] = None) -> int:
    parser = create_parser()
    args = parser.parse_args(argv)
    logging.basicConfig(level=args.log_level,
                        format="torchx %(levelname)s %(message)s")
    if not getattr(args, "func", None):
        parser.print_help()
        return 1
    return args.func(args) or 0


if __name__ == "__main__":
    sys.exit(main())

"""Utility components (parity: torchx/components/utils.py)."""

from __future__ import annotations

import shlex
from typing import Dict, List, Optional

import torchx_amd.specs as specs


def echo(msg: str = "hello world", image: str = specs.macros.img_root,
         num_replicas: int = 1) -> specs.AppDef:
    """Echos a message to stdout (for testing launch plumbing).

    Args:
        msg: message to echo
        image: image to use
        num_replicas: number of replicas
    """
    return specs.AppDef(
        name="echo",
        roles=[
            specs.Role(
                name="echo",
                image=image,
                entrypoint="echo",
                args=[msg],
                num_replicas=num_replicas,
                resource=specs.resource(),
            )
        ],
    )


def touch(file: str, image: str = specs.macros.img_root) -> specs.AppDef:
    """Touches a file.

    Args:
        file: file to create
        image: image to use
    """
    return specs.AppDef(
        name="touch",
        roles=[
            specs.Role(
                name="touch",
                image=image,
                entrypoint="touch",
                args=[file],
                resource=specs.resource(),
            )
        ],
    )


def sh(*args: str, image: str = specs.macros.img_root, num_replicas: int = 1,
       cpu: int = 2, gpu: int = 0, memMB: int = 1024,
       h: Optional[str] = None, env: Optional[Dict[str, str]] = None,
       max_retries: int = 0, mounts: Optional[List[str]] = None) -> specs.AppDef:
    """Runs the provided command via sh.

    Args:
        args: the command to run
        image: image to use
        num_replicas: replica count
        cpu: cpus per replica
        gpu: gpus per replica
        memMB: memory per replica
        h: named resource
        env: environment variables
        max_retries: scheduler retries
        mounts: mount specs
    """
    escaped = " ".join(shlex.quote(a) for a in args)
    return specs.AppDef(
        name="sh",
        roles=[
            specs.Role(
                name="sh",
                image=image,
                entrypoint="sh",
                args=["-c", escaped],
                env=env or {},
                num_replicas=num_replicas,
                max_retries=max_retries,
                resource=specs.resource(cpu=cpu, gpu=gpu, memMB=memMB, h=h),
                mounts=specs.parse_mounts(mounts) if mounts else [],
            )
        ],
    )


def python(*args: str, m: Optional[str] = None, c: Optional[str] = None,
           script: Optional[str] = None,
           image: str = specs.macros.img_root, name: str = "torchx_utils_python",
           cpu: int = 2, gpu: int = 0, memMB: int = 1024,
           h: Optional[str] = None, num_replicas: int = 1) -> specs.AppDef:
    """Runs python with the specified module, command or script.

    Args:
        args: arguments passed to the program
        m: run a module
        c: run a program string
        script: run a python script file
        image: image to use
        name: app name
        cpu: cpus per replica
        gpu: gpus per replica
        memMB: memory per replica
        h: named resource
        num_replicas: replica count
    """
    if sum(x is not None for x in (m, c, script)) != 1:
        raise ValueError("exactly one of m/c/script must be set")
    if m is not None:
        prog = ["-m", m]
    elif c is not None:
        prog = ["-c", c]
    else:
        prog = [script]  # type: ignore[list-item]
    return specs.AppDef(
        name=name,
        roles=[
            specs.Role(
                name=name,
                image=image,
                entrypoint="python3",
                args=["-u", *prog, *args],
                num_replicas=num_replicas,
                resource=specs.resource(cpu=cpu, gpu=gpu, memMB=memMB, h=h),
            )
        ],
    )


def binary(*args: str, entrypoint: str, name: str = "torchx_utils_binary",
           num_replicas: int = 1, cpu: int = 2, gpu: int = 0,
           memMB: int = 1024, h: Optional[str] = None) -> specs.AppDef:
    """Runs a prebuilt binary.

    Args:
        args: arguments to the binary
        entrypoint: the binary to run
        name: app name
        num_replicas: replica count
        cpu: cpus per replica
        gpu: gpus per replica
        memMB: memory per replica
        h: named resource
    """
    return specs.AppDef(
        name=name,
        roles=[
            specs.Role(
                name=name,
                image="<NONE>",
                entrypoint=entrypoint,
                args=list(args),
                num_replicas=num_replicas,
                resource=specs.resource(cpu=cpu, gpu=gpu, memMB=memMB, h=h),
            )
        ],
    )


def copy(src: str, dst: str, image: str = specs.macros.img_root) -> specs.AppDef:
    """Copies src to dst (fsspec URLs supported).

    Args:
        src: source path/url
        dst: destination path/url
        image: image to use
    """
    return specs.AppDef(
        name="copy",
        roles=[
            specs.Role(
                name="copy",
                image=image,
                entrypoint="python3",
                args=["-m", "torchx_amd.apps.copy_main", "--src", src,
                      "--dst", dst],
                resource=specs.resource(),
            )
        ],
    )


def booth(x1: float, x2: float, trial_idx: int = 0,
          tracker_base: str = "/tmp/torchx-booth",
          image: str = specs.macros.img_root) -> specs.AppDef:
    """Evaluates the booth function (HPO test objective).

    Args:
        x1: x1 value
        x2: x2 value
        trial_idx: trial index
        tracker_base: result tracker base path
        image: image to use
    """
    return specs.AppDef(
        name="booth",
        roles=[
            specs.Role(
                name="booth",
                image=image,
                entrypoint="python3",
                args=["-m", "torchx_amd.apps.booth_main", "--x1", str(x1),
                      "--x2", str(x2), "--trial_idx", str(trial_idx),
                      "--tracker_base", tracker_base],
                resource=specs.resource(),
            )
        ],
    )


def from_yaml(config: str, *overrides: str) -> specs.AppDef:
    """Build an AppDef from a YAML config file (parity with the reference's
    ``utils.hydra`` component, torchx/components/utils.py:330 — implemented
    dependency-free on plain YAML + dotted-path overrides).

    The file must have an ``app`` key describing the AppDef; ``role``
    entries are dicts of Role fields.  TorchX macros are available as
    ``${torchx.app_id}``, ``${torchx.replica_id}``, ``${torchx.rank0_env}``,
    ``${torchx.img_root}``.

    Args:
        config: path to the YAML config file
        overrides: dotted-path overrides, e.g. ``app.roles.0.num_replicas=2``
    """
    import yaml

    with open(config) as f:
        text = f.read()
    for name, macro in (
        ("app_id", specs.macros.app_id),
        ("replica_id", specs.macros.replica_id),
        ("rank0_env", specs.macros.rank0_env),
        ("img_root", specs.macros.img_root),
    ):
        text = text.replace("${torchx.%s}" % name, macro)
    cfg = yaml.safe_load(text)
    if not isinstance(cfg, dict) or "app" not in cfg:
        raise ValueError(f"{config} must contain a top-level `app` key")

    for ov in overrides:
        path, _, value = ov.partition("=")
        keys = path.split(".")
        node = cfg
        for k in keys[:-1]:
            node = node[int(k)] if isinstance(node, list) else node[k]
        leaf = keys[-1]
        parsed = yaml.safe_load(value)
        if isinstance(node, list):
            node[int(leaf)] = parsed
        else:
            node[leaf] = parsed

    app = cfg["app"]
    roles = []
    for r in app.get("roles", []):
        res = r.pop("resource", None)
        resource = (
            specs.Resource(**res) if isinstance(res, dict)
            else specs.named_resources()[res]() if isinstance(res, str)
            else specs.NULL_RESOURCE
        )
        roles.append(specs.Role(resource=resource, **r))
    return specs.AppDef(
        name=app.get("name", "app"),
        roles=roles,
        metadata=app.get("metadata", {}),
    )

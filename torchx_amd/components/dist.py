"""Distributed components (parity: torchx/components/dist.py).

``ddp`` builds a single-role AppDef whose entrypoint launches the
torchx_amd elastic agent (our native torchrun replacement) on every
replica, with a c10d rendezvous at the scheduler-provided rank-0 host
(``$TORCHX_RANK0_HOST``).  Workers bring up RCCL over xGMI.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import torchx_amd.specs as specs

_DEFAULT_RDZV_PORT = 29500

# RCCL/ROCm debug preset (reference _TORCH_DEBUG_FLAGS, dist.py:71 — the
# CUDA/NCCL names map to their ROCm equivalents: RCCL honours NCCL_* envs,
# AMD_LOG_LEVEL/HIP_LAUNCH_BLOCKING are the HIP-side switches)
_DEBUG_FLAGS: Dict[str, str] = {
    "HIP_LAUNCH_BLOCKING": "1",
    "AMD_LOG_LEVEL": "3",
    "NCCL_DEBUG": "INFO",
    "NCCL_DESYNC_DEBUG": "1",
    "TORCH_DISTRIBUTED_DEBUG": "DETAIL",
    "TORCH_SHOW_CPP_STACKTRACES": "1",
}


def parse_nnodes(j: str) -> Tuple[int, int, int, str]:
    """``j`` = "N", "NxM", or "minN:maxNxM" -> (min, max, nproc, nnodes_spec)."""
    if "x" in j:
        nodes, nproc = j.split("x")
    else:
        nodes, nproc = j, "1"
    if ":" in nodes:
        lo, hi = nodes.split(":")
    else:
        lo = hi = nodes
    return int(lo), int(hi), int(nproc), f"{lo}:{hi}"


def ddp(
    *script_args: str,
    script: Optional[str] = None,
    m: Optional[str] = None,
    image: str = specs.macros.img_root,
    name: str = "/",
    h: Optional[str] = None,
    cpu: int = 2,
    gpu: int = 0,
    memMB: int = 1024,
    j: str = "1x2",
    env: Optional[Dict[str, str]] = None,
    max_retries: int = 0,
    rdzv_port: int = _DEFAULT_RDZV_PORT,
    rdzv_backend: str = "c10d",
    mounts: Optional[List[str]] = None,
    debug: bool = False,
    tee: bool = True,
    rocprof: bool = False,
) -> specs.AppDef:
    """Distributed data-parallel application (elastic, one agent per node).

    Args:
        script_args: arguments passed to the user script/module
        script: python script to run (mutually exclusive with m)
        m: python module to run as ``python -m``
        image: image root (scheduler-dependent)
        name: ``experiment/run`` app name (``/`` -> derive from script)
        h: named resource (e.g. mi355x.8gpu); overrides cpu/gpu/memMB
        cpu: cpus per node
        gpu: gpus per node
        memMB: host memory per node
        j: ``{min_nnodes:}nnodes x nproc_per_node`` (e.g. 1x8, 1:4x8)
        env: extra environment variables
        max_retries: elastic agent max re-rendezvous restarts
        rdzv_port: c10d rendezvous port on the rank-0 host
        rdzv_backend: rendezvous backend (c10d)
        mounts: mount specs (type=bind,src=...,dst=...[,readonly])
        debug: enable the RCCL/ROCm debug env preset
        tee: prefix worker output with rank labels
        rocprof: wrap each worker in `rocprofv3 --kernel-trace --stats`
            (profiles land in $PET_LOG_DIR; SURVEY §5.1 launcher->profiling
            bridge)
    """
    if (script is None) == (m is None):
        raise ValueError("exactly one of --script / -m must be set")
    # allow the conventional `--` separator before script args
    if script_args and script_args[0] == "--":
        script_args = script_args[1:]

    min_n, max_n, nproc, nnodes_spec = parse_nnodes(j)

    app_name = name
    if name == "/" or not name:
        target = script or m or "ddp"
        app_name = os.path.splitext(os.path.basename(target))[0]
    elif "/" in name:
        exp, _, run = name.partition("/")
        app_name = run or exp

    # multi-node: rendezvous at the scheduler-provided rank-0 host; single
    # node: localhost (the app then also runs outside a scheduler —
    # reference trick, dist.py:231-243)
    endpoint = (
        f"${{TORCHX_RANK0_HOST}}:{rdzv_port}" if max_n > 1
        else f"localhost:{rdzv_port}"
    )

    agent_args: List[str] = [
        "-m", "torchx_amd.agent",
        "--nnodes", nnodes_spec,
        "--nproc-per-node", str(nproc),
        "--rdzv-endpoint", endpoint,
        "--rdzv-id", specs.macros.app_id,
        "--max-restarts", str(max_retries),
    ]
    if not tee:
        agent_args.append("--no-tee")
    if rocprof:
        agent_args.append("--rocprof")
    if m is not None:
        # agent execs `python3 -m <module>` via --no-python
        agent_args += ["--no-python", "python3", "-m", m]
    else:
        agent_args += [script]
    agent_args += list(script_args)

    role_env = {"LOGLEVEL": os.getenv("LOGLEVEL", "INFO")}
    if debug:
        role_env.update(_DEBUG_FLAGS)
    if env:
        role_env.update(env)

    return specs.AppDef(
        name=app_name,
        roles=[
            specs.Role(
                name=app_name,
                image=image,
                entrypoint="python3",
                args=agent_args,
                env=role_env,
                num_replicas=max_n,
                min_replicas=min_n if min_n != max_n else None,
                max_retries=max_retries,
                retry_policy=specs.RetryPolicy.REPLICA,
                resource=specs.resource(cpu=cpu, gpu=gpu, memMB=memMB, h=h),
                port_map={"c10d": rdzv_port},
                mounts=specs.parse_mounts(mounts) if mounts else [],
            )
        ],
    )


def spmd(
    *script_args: str,
    script: Optional[str] = None,
    m: Optional[str] = None,
    image: str = specs.macros.img_root,
    name: str = "/",
    h: str = "mi355x.8gpu",
    j: str = "1",
    env: Optional[Dict[str, str]] = None,
    max_retries: int = 0,
) -> specs.AppDef:
    """Single-program multi-data launch: like ddp but infers nproc_per_node
    from the named host's GPU count (parity: dist.py:87)."""
    if "x" in j:
        raise ValueError(
            f"j={j}: spmd takes node count only; nproc is inferred from -h"
        )
    ngpu = specs.resource(h=h).gpu or 1
    return ddp(
        *script_args, script=script, m=m, image=image, name=name, h=h,
        j=f"{j}x{ngpu}", env=env, max_retries=max_retries,
    )

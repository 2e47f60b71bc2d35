"""HIP/MI355X device enumeration and per-replica pinning.

The MI355X analog of the reference's `nvidia-smi -L` + CUDA_VISIBLE_DEVICES
partitioning (torchx/schedulers/local_scheduler.py:837-950): enumerate via
`amd-smi`/`rocm-smi` (torch.cuda as a fallback probe), partition contiguous
device-index ranges per replica, and export BOTH `HIP_VISIBLE_DEVICES` and
`ROCR_VISIBLE_DEVICES`.  Also maps named devices (e.g. RDMA NICs) to
DeviceMounts for containerized schedulers.
"""

from __future__ import annotations

import functools
import warnings
import json
import logging
import shutil
import subprocess
from typing import Any, Dict, List, Optional

from torchx_amd.specs import DeviceMount, Resource

log = logging.getLogger(__name__)

HIP_VISIBLE_DEVICES = "HIP_VISIBLE_DEVICES"
ROCR_VISIBLE_DEVICES = "ROCR_VISIBLE_DEVICES"

# named-device -> container device mounts (KFD + DRI are what ROCm containers
# need; see the docker scheduler)
def _amd_gpu_mounts(n: int) -> List[DeviceMount]:
    # /dev/kfd + /dev/dri expose every GPU; per-device isolation is done
    # with HIP_VISIBLE_DEVICES, not per-node mounts
    return [DeviceMount(src_path="/dev/kfd", dst_path="/dev/kfd"),
            DeviceMount(src_path="/dev/dri", dst_path="/dev/dri")]


def _efa_mounts(n: int) -> List[DeviceMount]:
    # parity: reference devices.py — one uverbs node per EFA device
    return [DeviceMount(src_path=f"/dev/infiniband/uverbs{i}",
                        dst_path=f"/dev/infiniband/uverbs{i}")
            for i in range(n)]


def _neuron_mounts(n: int) -> List[DeviceMount]:
    return [DeviceMount(src_path=f"/dev/neuron{i}",
                        dst_path=f"/dev/neuron{i}") for i in range(n)]


DEVICES: Dict[str, Any] = {
    "amd.com/gpu": _amd_gpu_mounts,
    "vpc.amazonaws.com/efa": _efa_mounts,
    "aws.amazon.com/neurondevice": _neuron_mounts,
}


def get_device_mounts(devices: Dict[str, int]) -> List[DeviceMount]:
    """Named device/count map -> DeviceMounts (parity: reference
    devices.py:43-54 — unknown names warn and are skipped)."""
    mounts: List[DeviceMount] = []
    for name, n in devices.items():
        fn = DEVICES.get(name)
        if fn is None:
            warnings.warn(f"could not find named device: {name}")
            continue
        mounts.extend(fn(int(n)))
    return mounts


@functools.lru_cache(maxsize=1)
def hip_device_count() -> int:
    """Number of visible AMD GPUs on this host (0 if none)."""
    # 1) amd-smi (ROCm >= 5.7)
    if shutil.which("amd-smi"):
        try:
            out = subprocess.run(
                ["amd-smi", "list", "--json"], capture_output=True, timeout=30
            )
            if out.returncode == 0:
                data = json.loads(out.stdout.decode() or "[]")
                if isinstance(data, list):
                    return len(data)
                if isinstance(data, dict):  # some versions nest under a key
                    for v in data.values():
                        if isinstance(v, list):
                            return len(v)
        except Exception as e:  # noqa: BLE001
            log.debug("amd-smi probe failed: %s", e)
    # 2) rocm-smi
    if shutil.which("rocm-smi"):
        try:
            out = subprocess.run(
                ["rocm-smi", "--showid", "--json"], capture_output=True,
                timeout=30,
            )
            if out.returncode == 0:
                data = json.loads(out.stdout.decode() or "{}")
                n = len([k for k in data if k.startswith("card")])
                if n:
                    return n
        except Exception as e:  # noqa: BLE001
            log.debug("rocm-smi probe failed: %s", e)
    # 3) torch runtime probe
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:  # noqa: BLE001
        pass
    return 0


def partition_devices(
    role_replicas: Dict[str, int],
    gpus_per_replica: Dict[str, int],
    total_gpus: Optional[int] = None,
) -> Dict[str, List[Optional[str]]]:
    """Assign contiguous device-index ranges per replica, role order.

    Returns role -> list (len num_replicas) of comma-joined device strings
    (None = don't set the env var, e.g. when over-subscribed).
    """
    if total_gpus is None:
        total_gpus = hip_device_count()
    assignments: Dict[str, List[Optional[str]]] = {}
    next_dev = 0
    requested = sum(
        role_replicas[r] * gpus_per_replica.get(r, 0) for r in role_replicas
    )
    oversubscribed = requested > total_gpus
    for role, n in role_replicas.items():
        per = gpus_per_replica.get(role, 0)
        out: List[Optional[str]] = []
        for _ in range(n):
            if per <= 0 or oversubscribed:
                out.append(None)
            else:
                out.append(",".join(str(d) for d in range(next_dev, next_dev + per)))
                next_dev += per
        assignments[role] = out
    if oversubscribed and requested > 0:
        log.warning(
            "requested %d GPUs but host has %d; not setting %s",
            requested, total_gpus, HIP_VISIBLE_DEVICES,
        )
    return assignments


def device_env(devices: Optional[str]) -> Dict[str, str]:
    if devices is None:
        return {}
    return {HIP_VISIBLE_DEVICES: devices, ROCR_VISIBLE_DEVICES: devices}


# ---------------------------------------------------------------------------
# NUMA affinity: pin each replica's CPU threads and page
# allocations to the NUMA node its GPUs hang off — the MI355X analog of the
# reference's CPU-affinity handling around local_scheduler.py:858-950.
# ---------------------------------------------------------------------------


def _read_device_numa_map(sysfs: str = "/sys") -> Dict[int, int]:
    """HIP device index -> NUMA node, from the DRM render nodes' sysfs
    (renderD* enumeration order matches the HIP device order)."""
    import glob
    import os
    import re

    result: Dict[int, int] = {}
    paths = sorted(
        glob.glob(os.path.join(sysfs, "class/drm/renderD*/device")),
        key=lambda p: int(re.search(r"renderD(\d+)", p).group(1)),
    )
    idx = 0
    for p in paths:
        try:
            with open(os.path.join(p, "vendor")) as f:
                vendor = f.read().strip()
        except OSError:
            continue
        if vendor != "0x1002":  # AMD
            continue
        try:
            with open(os.path.join(p, "numa_node")) as f:
                numa = int(f.read().strip())
        except (OSError, ValueError):
            numa = -1
        result[idx] = numa
        idx += 1
    return result


@functools.lru_cache(maxsize=1)
def device_numa_map() -> Dict[int, int]:
    return _read_device_numa_map()


def numa_node_of(devices: Optional[str],
                 numa_map: Optional[Dict[int, int]] = None) -> int:
    """The single NUMA node a device string ("0" / "0,1") sits on, or -1
    (unknown / spans nodes / no assignment)."""
    if not devices:
        return -1
    if numa_map is None:
        numa_map = device_numa_map()
    try:
        nodes = {numa_map.get(int(d), -1) for d in devices.split(",")}
    except ValueError:
        return -1
    if len(nodes) != 1:
        return -1
    return nodes.pop()


def numa_cpulist(node: int, sysfs: str = "/sys") -> str:
    """The kernel's cpulist string for a NUMA node ("0-31,64-95"), or ""."""
    if node < 0:
        return ""
    try:
        with open(f"{sysfs}/devices/system/node/node{node}/cpulist") as f:
            return f.read().strip()
    except OSError:
        return ""


def parse_cpulist(spec: str) -> set:
    """"0-3,8,10-11" -> {0,1,2,3,8,10,11}."""
    cpus: set = set()
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        lo, _, hi = part.partition("-")
        if hi:
            cpus.update(range(int(lo), int(hi) + 1))
        else:
            cpus.add(int(lo))
    return cpus


def numa_bind_args(
    devices: Optional[str],
    numa_map: Optional[Dict[int, int]] = None,
    numactl: Optional[str] = None,
) -> List[str]:
    """``numactl`` argv prefix binding a replica to its GPUs' NUMA node.

    Empty when: no device assignment, the devices span several nodes, the
    node is unknown (-1), or numactl is not installed.
    """
    if not devices:
        return []
    if numa_map is None:
        numa_map = device_numa_map()
    try:
        nodes = {numa_map.get(int(d), -1) for d in devices.split(",")}
    except ValueError:
        return []
    if len(nodes) != 1:
        return []
    node = nodes.pop()
    if node < 0:
        return []
    numactl = numactl if numactl is not None else shutil.which("numactl")
    if not numactl:
        return []
    return [numactl, f"--cpunodebind={node}", f"--membind={node}"]

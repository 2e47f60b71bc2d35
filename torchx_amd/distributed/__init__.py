"""In-job helpers over torch.distributed (parity:
torchx/distributed/__init__.py:27-305).

``init_pg(backend="auto")`` picks RCCL (PyTorch-ROCm names it "nccl") when
HIP GPUs are present, else gloo; outside a launcher it creates a trivial
single-process group.
"""

from __future__ import annotations

import os
from contextlib import contextmanager
from typing import Iterator, Optional

import torch
import torch.distributed as dist


def is_torchelastic_launched() -> bool:
    return "TORCHELASTIC_RUN_ID" in os.environ or (
        "RANK" in os.environ and "WORLD_SIZE" in os.environ
    )


def rank() -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank()
    return int(os.environ.get("RANK", "0"))


def world_size() -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get("WORLD_SIZE", "1"))


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", "0"))


def local_device() -> torch.device:
    if torch.cuda.is_available():
        # modulo: more local ranks than devices = ranks sharing a GPU
        # (single-GPU multi-rank RCCL tests); 1:1 in production
        return torch.device("cuda", local_rank() % torch.cuda.device_count())
    return torch.device("cpu")


def _auto_backend() -> str:
    if (
        torch.cuda.is_available()
        and torch.cuda.device_count() > 0
        and dist.is_nccl_available()
    ):
        return "nccl"  # RCCL on ROCm
    return "gloo"


def _apply_cpu_affinity() -> None:
    """Pin this worker to its GPU's NUMA-node CPUs if the agent exported
    TORCHX_AMD_CPU_AFFINITY (the numactl-free affinity path)."""
    spec = os.environ.get("TORCHX_AMD_CPU_AFFINITY")
    if not spec:
        return
    try:
        from torchx_amd.schedulers.devices import parse_cpulist

        cpus = parse_cpulist(spec)
        if cpus:
            os.sched_setaffinity(0, cpus)
    except (OSError, ValueError):  # affinity is best-effort
        pass


def init_pg(backend: str = "auto") -> torch.device:
    """Initialize the default process group and return this rank's device."""
    _apply_cpu_affinity()
    if backend == "auto":
        backend = _auto_backend()
    if not dist.is_initialized():
        if not is_torchelastic_launched():
            # trivial single-process group
            os.environ.setdefault("MASTER_ADDR", "localhost")
            os.environ.setdefault("MASTER_PORT", "0")
            os.environ.setdefault("RANK", "0")
            os.environ.setdefault("WORLD_SIZE", "1")
            store = dist.TCPStore(
                "localhost", 0, is_master=True, wait_for_workers=False
            )
            dist.init_process_group(backend, store=store, rank=0, world_size=1)
        else:
            dist.init_process_group(backend)
    dev = local_device()
    if dev.type == "cuda":
        torch.cuda.set_device(dev)
    return dev


@contextmanager
def on_rank0_first() -> Iterator[None]:
    """Rank 0 runs the body before everyone else (e.g. dataset download)."""
    if world_size() > 1 and rank() != 0:
        dist.barrier()
    yield
    if world_size() > 1 and rank() == 0:
        dist.barrier()


@contextmanager
def on_local_rank0_first() -> Iterator[None]:
    if world_size() > 1 and local_rank() != 0:
        dist.barrier()
    yield
    if world_size() > 1 and local_rank() == 0:
        dist.barrier()

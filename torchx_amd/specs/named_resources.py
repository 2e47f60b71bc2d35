"""Named resource shapes, MI355X-first.

``mi355x.1gpu`` .. ``mi355x.8gpu`` size host cpu/mem for slices of a standard
8x MI355X node (2 socket EPYC, ~192 cores / ~2.3 TB host RAM class), with
288 GB HBM3E per GPU recorded as a capability.  Generic ``gpu.small`` /
``cpu.*`` t-shirt sizes are kept for parity with the reference
(torchx/specs/named_resources_generic.py:47).
"""

from __future__ import annotations

from typing import Callable, Dict

from .api import Resource

# One 8xMI355X node: leave ~4% host-mem tax for the OS/daemons (same idea as
# the reference's AWS MEM_TAX, named_resources_aws.py:47).
_NODE_CPU = 192
_NODE_MEM_MB = int(2_304 * 1024 * 0.96)
HBM_GB_PER_GPU = 288


def _mi355x(num_gpus: int) -> Callable[[], Resource]:
    def factory() -> Resource:
        from .capabilities import GFX_ARCH, HBM_GB, XGMI_LINKS

        frac = num_gpus / 8
        r = Resource(
            cpu=int(_NODE_CPU * frac),
            gpu=num_gpus,
            memMB=int(_NODE_MEM_MB * frac),
            capabilities={"amd.com/gpu_arch": "gfx950",
                          "amd.com/hbm_gb": HBM_GB_PER_GPU},
        )
        # typed capability keys (specs/capabilities.py) for scheduler-side
        # topology decisions: 7 point-to-point xGMI links per MI355X
        GFX_ARCH.set(r, "gfx950")
        HBM_GB.set(r, HBM_GB_PER_GPU)
        XGMI_LINKS.set(r, 7)
        return r

    return factory


def _generic(cpu: int, gpu: int, memMB: int) -> Callable[[], Resource]:
    return lambda: Resource(cpu=cpu, gpu=gpu, memMB=memMB)


NAMED_RESOURCES: Dict[str, Callable[[], Resource]] = {
    # MI355X-native shapes
    "mi355x.1gpu": _mi355x(1),
    "mi355x.2gpu": _mi355x(2),
    "mi355x.4gpu": _mi355x(4),
    "mi355x.8gpu": _mi355x(8),
    # generic t-shirts (parity with reference generic shapes)
    "gpu.small": _generic(8, 1, 56 * 1024),
    "gpu.medium": _generic(16, 2, 112 * 1024),
    "gpu.large": _generic(32, 4, 224 * 1024),
    "gpu.xlarge": _generic(64, 8, 448 * 1024),
    "cpu.nano": _generic(1, 0, 2 * 1024),
    "cpu.micro": _generic(2, 0, 4 * 1024),
    "cpu.small": _generic(4, 0, 8 * 1024),
    "cpu.medium": _generic(8, 0, 16 * 1024),
    "cpu.large": _generic(16, 0, 32 * 1024),
    "cpu.xlarge": _generic(32, 0, 64 * 1024),
}

"""Cloud-instance named resources (parity: torchx/specs/named_resources_aws.py).

The MI355X-native library lives in ``named_resources.py`` (``mi355x.*``);
this module keeps the reference's cloud vocabulary working so existing
torchx workflows resolve: common AWS GPU/CPU shapes with the reference's
conventions — 96% memory tax (host daemons) and the
``node.kubernetes.io/instance-type`` capability for k8s node selection.
"""

from __future__ import annotations

from typing import Callable, Dict

from .api import Resource

MEM_TAX = 0.96
K8S_ITYPE = "node.kubernetes.io/instance-type"
GiB = 1024


def _aws(name: str, cpu: int, gpu: int, mem_gib: int,
         devices: Dict[str, int] = None) -> Callable[[], Resource]:
    def factory() -> Resource:
        return Resource(
            cpu=cpu, gpu=gpu, memMB=int(mem_gib * GiB * MEM_TAX),
            capabilities={K8S_ITYPE: name},
            devices=dict(devices or {}),
        )

    return factory


EFA = "vpc.amazonaws.com/efa"

NAMED_RESOURCES: Dict[str, Callable[[], Resource]] = {
    # GPU trainers
    "aws_p3.2xlarge": _aws("p3.2xlarge", 8, 1, 61),
    "aws_p3.8xlarge": _aws("p3.8xlarge", 32, 4, 244),
    "aws_p3.16xlarge": _aws("p3.16xlarge", 64, 8, 488),
    "aws_p3dn.24xlarge": _aws("p3dn.24xlarge", 96, 8, 768, {EFA: 1}),
    "aws_p4d.24xlarge": _aws("p4d.24xlarge", 96, 8, 1152, {EFA: 4}),
    "aws_p4de.24xlarge": _aws("p4de.24xlarge", 96, 8, 1152, {EFA: 4}),
    "aws_p5.48xlarge": _aws("p5.48xlarge", 192, 8, 2048, {EFA: 32}),
    "aws_g4dn.xlarge": _aws("g4dn.xlarge", 4, 1, 16),
    "aws_g4dn.12xlarge": _aws("g4dn.12xlarge", 48, 4, 192),
    "aws_g5.xlarge": _aws("g5.xlarge", 4, 1, 16),
    "aws_g5.12xlarge": _aws("g5.12xlarge", 48, 4, 192),
    "aws_g5.48xlarge": _aws("g5.48xlarge", 192, 8, 768),
    "aws_g6e.12xlarge": _aws("g6e.12xlarge", 48, 4, 384),
    # CPU hosts
    "aws_m5.large": _aws("m5.large", 2, 0, 8),
    "aws_m5.2xlarge": _aws("m5.2xlarge", 8, 0, 32),
    "aws_m5.4xlarge": _aws("m5.4xlarge", 16, 0, 64),
    "aws_c5.4xlarge": _aws("c5.4xlarge", 16, 0, 32),
    "aws_t3.medium": _aws("t3.medium", 2, 0, 4),
}

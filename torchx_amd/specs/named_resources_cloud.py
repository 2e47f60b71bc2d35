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
NEURON = "aws.amazon.com/neurondevice"

NAMED_RESOURCES: Dict[str, Callable[[], Resource]] = {
    # GPU trainers (P family)
    "aws_p3.2xlarge": _aws("p3.2xlarge", 8, 1, 61),
    "aws_p3.8xlarge": _aws("p3.8xlarge", 32, 4, 244),
    "aws_p3.16xlarge": _aws("p3.16xlarge", 64, 8, 488),
    "aws_p3dn.24xlarge": _aws("p3dn.24xlarge", 96, 8, 768, {EFA: 1}),
    "aws_p4d.24xlarge": _aws("p4d.24xlarge", 96, 8, 1152, {EFA: 4}),
    "aws_p4de.24xlarge": _aws("p4de.24xlarge", 96, 8, 1152, {EFA: 4}),
    "aws_p5.48xlarge": _aws("p5.48xlarge", 192, 8, 2048, {EFA: 32}),
    "aws_p5e.48xlarge": _aws("p5e.48xlarge", 192, 8, 2048, {EFA: 32}),
    "aws_p5en.48xlarge": _aws("p5en.48xlarge", 192, 8, 2048, {EFA: 16}),
    "aws_p6-b200.48xlarge": _aws("p6-b200.48xlarge", 192, 8, 2048, {EFA: 8}),
    "aws_p6-b300.48xlarge": _aws("p6-b300.48xlarge", 192, 8, 4096, {EFA: 16}),
    "aws_p6e-gb200.36xlarge": _aws("p6e-gb200.36xlarge", 144, 4, 960,
                                   {EFA: 16}),
    # GPU inference/graphics (G family)
    "aws_g4dn.xlarge": _aws("g4dn.xlarge", 4, 1, 16),
    "aws_g4dn.2xlarge": _aws("g4dn.2xlarge", 8, 1, 32),
    "aws_g4dn.4xlarge": _aws("g4dn.4xlarge", 16, 1, 64),
    "aws_g4dn.8xlarge": _aws("g4dn.8xlarge", 32, 1, 128, {EFA: 1}),
    "aws_g4dn.12xlarge": _aws("g4dn.12xlarge", 48, 4, 192, {EFA: 1}),
    "aws_g4dn.16xlarge": _aws("g4dn.16xlarge", 64, 1, 256, {EFA: 1}),
    "aws_g4dn.metal": _aws("g4dn.metal", 96, 8, 384, {EFA: 1}),
    "aws_g5.xlarge": _aws("g5.xlarge", 4, 1, 16),
    "aws_g5.2xlarge": _aws("g5.2xlarge", 8, 1, 32),
    "aws_g5.4xlarge": _aws("g5.4xlarge", 16, 1, 64),
    "aws_g5.8xlarge": _aws("g5.8xlarge", 32, 1, 128, {EFA: 1}),
    "aws_g5.12xlarge": _aws("g5.12xlarge", 48, 4, 192, {EFA: 1}),
    "aws_g5.16xlarge": _aws("g5.16xlarge", 64, 1, 256, {EFA: 1}),
    "aws_g5.24xlarge": _aws("g5.24xlarge", 96, 4, 384, {EFA: 1}),
    "aws_g5.48xlarge": _aws("g5.48xlarge", 192, 8, 768, {EFA: 1}),
    "aws_g6e.xlarge": _aws("g6e.xlarge", 4, 1, 32),
    "aws_g6e.2xlarge": _aws("g6e.2xlarge", 8, 1, 64),
    "aws_g6e.4xlarge": _aws("g6e.4xlarge", 16, 1, 128),
    "aws_g6e.8xlarge": _aws("g6e.8xlarge", 32, 1, 256),
    "aws_g6e.12xlarge": _aws("g6e.12xlarge", 48, 4, 384),
    "aws_g6e.16xlarge": _aws("g6e.16xlarge", 64, 1, 512),
    "aws_g6e.24xlarge": _aws("g6e.24xlarge", 96, 4, 768, {EFA: 2}),
    "aws_g6e.48xlarge": _aws("g6e.48xlarge", 192, 8, 1536, {EFA: 4}),
    # Trainium / Inferentia
    "aws_trn1.2xlarge": _aws("trn1.2xlarge", 8, 0, 32, {NEURON: 1}),
    "aws_trn1.32xlarge": _aws("trn1.32xlarge", 128, 0, 512,
                              {EFA: 8, NEURON: 16}),
    "aws_inf2.xlarge": _aws("inf2.xlarge", 4, 0, 16, {NEURON: 1}),
    "aws_inf2.8xlarge": _aws("inf2.8xlarge", 32, 0, 128, {NEURON: 1}),
    "aws_inf2.24xlarge": _aws("inf2.24xlarge", 96, 0, 384, {NEURON: 6}),
    "aws_inf2.48xlarge": _aws("inf2.48xlarge", 192, 0, 768, {NEURON: 12}),
    # CPU hosts (M/C/T families)
    "aws_m5.large": _aws("m5.large", 2, 0, 8),
    "aws_m5.xlarge": _aws("m5.xlarge", 4, 0, 16),
    "aws_m5.2xlarge": _aws("m5.2xlarge", 8, 0, 32),
    "aws_m5.4xlarge": _aws("m5.4xlarge", 16, 0, 64),
    "aws_m5.8xlarge": _aws("m5.8xlarge", 32, 0, 128),
    "aws_m5.12xlarge": _aws("m5.12xlarge", 48, 0, 192),
    "aws_m5.16xlarge": _aws("m5.16xlarge", 64, 0, 256),
    "aws_m5.24xlarge": _aws("m5.24xlarge", 96, 0, 384),
    "aws_m5.metal": _aws("m5.metal", 96, 0, 384),
    "aws_m5d.large": _aws("m5d.large", 2, 0, 8),
    "aws_m5d.xlarge": _aws("m5d.xlarge", 4, 0, 16),
    "aws_m5d.2xlarge": _aws("m5d.2xlarge", 8, 0, 32),
    "aws_m5d.4xlarge": _aws("m5d.4xlarge", 16, 0, 64),
    "aws_m5d.8xlarge": _aws("m5d.8xlarge", 32, 0, 128),
    "aws_m5d.12xlarge": _aws("m5d.12xlarge", 48, 0, 192),
    "aws_m5d.16xlarge": _aws("m5d.16xlarge", 64, 0, 256),
    "aws_m5d.24xlarge": _aws("m5d.24xlarge", 96, 0, 384),
    "aws_m5d.metal": _aws("m5d.metal", 96, 0, 384),
    "aws_c5.4xlarge": _aws("c5.4xlarge", 16, 0, 32),
    "aws_c5.18xlarge": _aws("c5.18xlarge", 72, 0, 142),
    "aws_t3.medium": _aws("t3.medium", 2, 0, 4),
}

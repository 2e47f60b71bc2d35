"""Structured component arguments: ``--name exp/run`` and ``-j NxM``.

Behavior parity with the reference (torchx/components/structured_arg.py:36,156):
``StructuredNameArgument`` splits an ``{experiment}/{run}`` name, deriving the
run name from the main module or script when omitted; ``StructuredJArgument``
parses ``-j nnodes[xnproc]``, inferring ``nproc_per_node`` from the named
host's GPU count (``-j 2`` on an 8-GPU ``mi355x.8gpu`` host means ``2x8``).
"""

from __future__ import annotations

import warnings
from dataclasses import dataclass
from pathlib import Path
from typing import Optional

from torchx_amd import specs


@dataclass
class StructuredNameArgument:
    experiment_name: str
    run_name: str

    def __str__(self) -> str:
        return f"{self.experiment_name or ''}/{self.run_name}"

    @staticmethod
    def parse_from(
        name: str,
        m: Optional[str] = None,
        script: Optional[str] = None,
        default_experiment_name: str = "default-experiment",
    ) -> "StructuredNameArgument":
        """Parse ``{experiment}/{run}``; either side may be empty.

        ``foo/`` names only the experiment (run derived from m/script stem),
        ``/bar`` or ``bar`` names only the run, ``foo/bar`` names both, and
        ``/`` derives the run and uses the default experiment.
        """
        if bool(m) == bool(script):
            raise ValueError(
                "specify exactly one of: main module (m) or script"
            )
        experiment_name = ""
        run_name = ""
        delim = name.find("/")
        if delim < 0:
            run_name = name
        elif delim < len(name) - 1:
            run_name = name[delim + 1:]
        if delim > 0:
            experiment_name = name[:delim]
        if not run_name:
            run_name = m.rpartition(".")[2] if m else Path(script).stem
        return StructuredNameArgument(
            experiment_name or default_experiment_name, run_name
        )


@dataclass
class StructuredJArgument:
    nnodes: int
    nproc_per_node: int

    def __str__(self) -> str:
        return f"{self.nnodes}x{self.nproc_per_node}"

    @staticmethod
    def parse_from(h: str, j: str) -> "StructuredJArgument":
        """Parse ``-j nnodes[xnproc]`` against named host ``h``.

        With only nnodes given, nproc_per_node is the host's GPU count
        (error if the host has none). An explicit nproc that mismatches the
        GPU count is honored with a warning.
        """
        nums = j.split("x")
        num_gpus = specs.named_resources()[h]().gpu
        if len(nums) == 1:
            nnodes = int(nums[0])
            if num_gpus <= 0:
                raise ValueError(
                    f"nproc_per_node cannot be inferred: `{h}` has no GPUs; "
                    f"specify `-j {nnodes}xN`"
                )
            nproc_per_node = num_gpus
        elif len(nums) == 2:
            nnodes = int(nums[0])
            nproc_per_node = int(nums[1])
            if nproc_per_node != num_gpus:
                warnings.warn(
                    f"-j {j}: nproc_per_node={nproc_per_node} != GPU count "
                    f"of {h} ({num_gpus}); this may under-utilize the host"
                )
        else:
            raise ValueError(f"invalid -j format (want NNODESxNPROC): {j}")
        return StructuredJArgument(nnodes=nnodes, nproc_per_node=nproc_per_node)

"""Fused flat-buffer AdamW with fp32 master weights.

One HIP kernel launch per weight-decay group per step (two total for a
whole Llama-3-8B), each a single HBM-rate sweep over {p32, p16, g, m, v}.
"""

from __future__ import annotations

from typing import Dict

import torch

from torchx_amd import ops
from .flat import FlatParams


class FlatAdamW:
    def __init__(
        self,
        flat: FlatParams,
        lr: float = 3e-4,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.1,
    ):
        self.flat = flat
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.state: Dict[str, Dict[str, torch.Tensor]] = {}
        for g, p16, _ in flat.groups():
            self.state[g] = {
                "p32": p16.float(),
                "m": torch.zeros_like(p16, dtype=torch.float32),
                "v": torch.zeros_like(p16, dtype=torch.float32),
            }

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        for g, p16, grad in self.flat.groups():
            st = self.state[g]
            wd = 0.0 if g == "no_decay" else self.weight_decay
            ops.adamw_step(
                st["p32"], p16, grad, st["m"], st["v"],
                lr=self.lr, beta1=self.beta1, beta2=self.beta2, eps=self.eps,
                weight_decay=wd, step=self.step_count,
            )

    def zero_grad(self) -> None:
        self.flat.zero_grad()

    def state_dict(self) -> Dict:
        return {
            "step": self.step_count,
            "lr": self.lr,
            "state": {g: {k: v for k, v in st.items()}
                      for g, st in self.state.items()},
        }

    def load_state_dict(self, sd: Dict) -> None:
        self.step_count = sd["step"]
        self.lr = sd["lr"]
        for g, st in sd["state"].items():
            for k, v in st.items():
                self.state[g][k].copy_(v)

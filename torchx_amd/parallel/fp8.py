"""fp8 (OCP e4m3/e5m2) linear layers over hipBLASLt scaled GEMMs.

MI355X's matrix cores run fp8 at ~2x the bf16 rate (measured 2.8 PF/s vs
1.48 PF/s on the fat MLP shape — tools/probe_fp8.py). ``Fp8Linear`` is a
drop-in for ``nn.Linear(bias=False)``: per-tensor dynamic scaling, e4m3
for activations/weights, e5m2 for gradients, all three GEMMs (fwd, dgrad,
wgrad) in fp8 via ``torch._scaled_mm``. The master weight stays a normal
bf16 Parameter, so FlatParams/FlatDDP/FlatAdamW work unchanged — fp8 is
purely a compute-path transform (``convert_to_fp8(model)`` before
FlatParams).

This is an OPT-IN mode (``bench.py --dtype fp8``); the headline benchmark
stays bf16.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def _amax_scale(t: torch.Tensor, fmax: float) -> torch.Tensor:
    amax = torch.amax(t.abs().float())
    return torch.clamp(amax / fmax, min=1e-12)


def _quant(t: torch.Tensor, scale: torch.Tensor,
           dtype: torch.dtype) -> torch.Tensor:
    return (t.float() / scale).to(dtype)


class _Fp8Matmul(torch.autograd.Function):
    """y = x @ W^T with all three GEMMs in fp8.

    Forward saves BOTH row orientations of the fp8 operands (the
    TransformerEngine recipe) so dgrad and wgrad feed ``_scaled_mm``
    directly — its B operand must be column-major.
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor):
        # x [M, K] bf16; w [N, K] bf16
        M, K = x.shape
        sx = _amax_scale(x, E4M3_MAX)
        sw = _amax_scale(w, E4M3_MAX)
        x8 = _quant(x, sx, torch.float8_e4m3fn)            # [M, K] row
        w8 = _quant(w, sw, torch.float8_e4m3fn)            # [N, K] row
        y = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                             out_dtype=torch.bfloat16)
        # transposed fp8 copies for backward
        x8t = _quant(x.t().contiguous(), sx, torch.float8_e4m3fn)  # [K, M]
        w8t = _quant(w.t().contiguous(), sw, torch.float8_e4m3fn)  # [K, N]
        ctx.save_for_backward(x8t, w8t, sx, sw)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x8t, w8t, sx, sw = ctx.saved_tensors
        dy = dy.contiguous()
        sg = _amax_scale(dy, E5M2_MAX)
        dy8 = _quant(dy, sg, torch.float8_e5m2)            # [M, N] row
        dy8t = _quant(dy.t().contiguous(), sg, torch.float8_e5m2)  # [N, M]
        # dx [M, K] = dy [M, N] @ W [N, K];  B col-major = w8t.t()
        dx = torch._scaled_mm(dy8, w8t.t(), scale_a=sg, scale_b=sw,
                              out_dtype=torch.bfloat16)
        # dW [N, K] = dy^T [N, M] @ x [M, K];  B col-major = x8t.t()
        dw = torch._scaled_mm(dy8t, x8t.t(), scale_a=sg, scale_b=sx,
                              out_dtype=torch.bfloat16)
        return dx, dw


class Fp8Linear(nn.Module):
    """Drop-in for ``nn.Linear(in, out, bias=False)`` with fp8 GEMMs."""

    def __init__(self, in_features: int, out_features: int,
                 weight: Optional[nn.Parameter] = None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        if weight is None:
            weight = nn.Parameter(
                torch.empty(out_features, in_features,
                            dtype=torch.bfloat16)
            )
            nn.init.normal_(weight, std=in_features ** -0.5)
        self.weight = weight

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape
        x2 = x.reshape(-1, self.in_features)
        if x.is_cuda:
            y = _Fp8Matmul.apply(x2, self.weight)
        else:  # CPU CI path: plain bf16 matmul
            y = x2 @ self.weight.t()
        return y.reshape(*shape[:-1], self.out_features)

    def extra_repr(self) -> str:
        return (f"in_features={self.in_features}, "
                f"out_features={self.out_features}, fp8=e4m3/e5m2")


def convert_to_fp8(module: nn.Module,
                   min_features: int = 1024) -> nn.Module:
    """Replace every bias-free ``nn.Linear`` whose dims are fp8-friendly
    (divisible by 16, at least ``min_features``) with an :class:`Fp8Linear`
    SHARING the same weight Parameter. Call BEFORE FlatParams so the
    flat-buffer views attach to the shared weights."""
    for name, child in list(module.named_children()):
        if (isinstance(child, nn.Linear) and child.bias is None
                and child.in_features % 16 == 0
                and child.out_features % 16 == 0
                and child.in_features >= min_features):
            repl = Fp8Linear(child.in_features, child.out_features,
                             weight=child.weight)
            setattr(module, name, repl)
        else:
            convert_to_fp8(child, min_features)
    return module

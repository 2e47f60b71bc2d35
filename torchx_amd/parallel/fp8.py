"""fp8 (OCP e4m3/e5m2) linear layers over hipBLASLt scaled GEMMs.

MI355X's matrix cores run fp8 at ~2x the bf16 rate (measured 2.8 PF/s vs
1.48 PF/s on the fat MLP shape — tools/probe_fp8.py). ``Fp8Linear`` is a
drop-in for ``nn.Linear(bias=False)``: all three GEMMs (fwd, dgrad, wgrad)
— all three GEMMs (fwd, dgrad, wgrad) run in e4m3 via
``torch._scaled_mm`` (the cast kernel clamps to the finite fp8 max first:
hardware overflow encodes NaN, which poisoned delayed scaling until
calibration). Quantization uses the fused HIP cast+transpose kernel
(ops/csrc/fp8_cast.hip): ONE read of the bf16 tensor yields both fp8
orientations (``_scaled_mm`` needs a column-major B) plus the amax for the
next step's scale (TransformerEngine-style delayed scaling — no host
syncs; the first optimizer step sees a warm-up scale seeded at conversion).

The master weight stays a normal bf16 Parameter, so FlatParams / FlatDDP /
FlatAdamW work unchanged — fp8 is purely a compute-path transform
(``convert_to_fp8(model)`` before FlatParams). Opt-in via
``bench.py --dtype fp8``; the headline benchmark stays bf16.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn

E4M3_MAX = 448.0
E5M2_MAX = 57344.0

# experiment knob for round 2: e5m2 gradients (wider range, less mantissa);
# the mixed e5m2 x e4m3 scaled-mm path needs re-validation now that the
# overflow-NaN poisoning is fixed
_E5M2_GRADS = os.environ.get("TORCHX_AMD_FP8_E5M2_GRADS", "0") == "1"
_GRAD_MAX = E5M2_MAX if _E5M2_GRADS else E4M3_MAX


def _hip():
    from torchx_amd import ops

    return ops.hip_ops(required=True)


def _scale_from_amax(amax: torch.Tensor, fmax: float) -> torch.Tensor:
    return torch.clamp(amax / fmax, min=1e-10)


class _Fp8Matmul(torch.autograd.Function):
    """y = x @ W^T, all GEMMs e4m3; fused cast+transpose, delayed scaling."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, mod: "Fp8Linear"):
        hip = _hip()
        sx = _scale_from_amax(mod.x_amax, E4M3_MAX)
        sw = _scale_from_amax(mod.w_amax, E4M3_MAX)
        x8, x8t, x_amax = hip.fp8_cast_transpose(x, sx, False)
        w8, w8t, w_amax = hip.fp8_cast_transpose(w, sw, False)
        mod.x_amax.copy_(x_amax[0])        # async: feeds the NEXT call
        mod.w_amax.copy_(w_amax[0])
        y = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                             out_dtype=torch.bfloat16)
        ctx.save_for_backward(x8t, w8t, sx, sw)
        ctx.mod = mod
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x8t, w8t, sx, sw = ctx.saved_tensors
        mod = ctx.mod
        hip = _hip()
        sg = _scale_from_amax(mod.g_amax, _GRAD_MAX)
        dy8, dy8t, g_amax = hip.fp8_cast_transpose(dy.contiguous(), sg,
                                                   _E5M2_GRADS)
        mod.g_amax.copy_(g_amax[0])
        # dx [M, K] = dy [M, N] @ W [N, K];  B col-major = w8t.t()
        dx = torch._scaled_mm(dy8, w8t.t(), scale_a=sg, scale_b=sw,
                              out_dtype=torch.bfloat16)
        # dW [N, K] = dy^T [N, M] @ x [M, K];  B col-major = x8t.t()
        dw = torch._scaled_mm(dy8t, x8t.t(), scale_a=sg, scale_b=sx,
                              out_dtype=torch.bfloat16)
        return dx, dw, None


class Fp8Linear(nn.Module):
    """Drop-in for ``nn.Linear(in, out, bias=False)`` with fp8 GEMMs."""

    def __init__(self, in_features: int, out_features: int,
                 weight: Optional[nn.Parameter] = None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        if weight is None:
            weight = nn.Parameter(
                torch.empty(out_features, in_features, dtype=torch.bfloat16)
            )
            nn.init.normal_(weight, std=in_features ** -0.5)
        self.weight = weight
        # delayed-scaling state: amax of the PREVIOUS step's tensors.
        # w seeded exactly at conversion; x/g warm up over the first steps
        # (cast saturates to the fp8 max meanwhile).
        dev = weight.device
        w_amax = weight.detach().abs().amax().float()
        self.register_buffer("w_amax", w_amax.clone())
        self.register_buffer("x_amax", torch.ones((), device=dev))
        self.register_buffer("g_amax", torch.ones((), device=dev))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape
        x2 = x.reshape(-1, self.in_features)
        if x.is_cuda:
            # the fused cast+transpose kernel stores 8-wide vectors: pad the
            # token dim up to a multiple of 8 (zero rows quantize to zero and
            # the slice below keeps their grads out of the graph)
            rows = x2.shape[0]
            pad = (-rows) % 8
            if pad:
                x2 = torch.nn.functional.pad(x2, (0, 0, 0, pad))
            y = _Fp8Matmul.apply(x2, self.weight, self)
            if pad:
                y = y[:rows]
        else:  # CPU CI path: plain bf16 matmul
            y = x2 @ self.weight.t()
        return y.reshape(*shape[:-1], self.out_features)

    def extra_repr(self) -> str:
        return (f"in_features={self.in_features}, "
                f"out_features={self.out_features}, fp8=e4m3(delayed)")


DEFAULT_FP8_EXCLUDE = ("lm_head", "output", "head", "embed")


def convert_to_fp8(module: nn.Module,
                   min_features: int = 1024,
                   exclude: tuple = DEFAULT_FP8_EXCLUDE) -> nn.Module:
    """Replace every bias-free ``nn.Linear`` whose dims are fp8-friendly
    (divisible by 16, at least ``min_features``) with an :class:`Fp8Linear`
    SHARING the same weight Parameter. Call BEFORE FlatParams so the
    flat-buffer views attach to the shared weights.

    Modules whose name contains any ``exclude`` substring stay in high
    precision — standard fp8 recipes (TransformerEngine) keep the output
    projection out of e4m3: quantized logits feeding cross-entropy degrade
    loss quality. Weights tied to an excluded module (e.g. lm_head sharing
    the embedding parameter) are also left alone.
    """
    # collect parameters owned by excluded modules so tied weights skip too
    excluded_params = set()
    for qname, sub in module.named_modules():
        leaf = qname.rsplit(".", 1)[-1]
        if any(pat in leaf for pat in exclude):
            for p in sub.parameters(recurse=True):
                excluded_params.add(id(p))

    def _convert(mod: nn.Module) -> None:
        for name, child in list(mod.named_children()):
            if (isinstance(child, nn.Linear) and child.bias is None
                    and child.in_features % 16 == 0
                    and child.out_features % 16 == 0
                    and child.in_features >= min_features
                    and not any(pat in name for pat in exclude)
                    and id(child.weight) not in excluded_params):
                repl = Fp8Linear(child.in_features, child.out_features,
                                 weight=child.weight)
                setattr(mod, name, repl)
            else:
                _convert(child)

    _convert(module)
    return module

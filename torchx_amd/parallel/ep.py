"""Expert-parallel token dispatch over RCCL all-to-all (xGMI-native).

The 8-GPU MI355X node is fully connected point-to-point (7 xGMI links per
GPU), so all-to-all is the natural EP transport: each pairwise exchange
rides its own link with no switch contention.  Dispatch:

  1. top-k routing -> sort tokens by destination expert
  2. exchange per-rank counts (small all_to_all)
  3. all_to_all_single of the hidden states (token payload)
  4. local expert compute (experts_per_rank = E / ep_size)
  5. reverse all_to_all, unpermute, weighted combine

A gloo-compatible fallback (all_gather + slicing) keeps the math testable
in CPU CI (gloo lacks all_to_all); the RCCL path is the production one.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def _all_to_all_single(
    out: torch.Tensor, inp: torch.Tensor,
    out_splits: List[int], in_splits: List[int],
    group: Optional[dist.ProcessGroup],
) -> None:
    backend = dist.get_backend(group)
    if backend == "nccl":  # RCCL
        dist.all_to_all_single(out, inp, out_splits, in_splits, group=group)
        return
    # gloo fallback: exchange via all_gather of (padded) buffers
    ws = dist.get_world_size(group)
    rank = dist.get_rank(group)
    max_n = torch.tensor([inp.shape[0]], dtype=torch.long)
    sizes = [torch.zeros_like(max_n) for _ in range(ws)]
    dist.all_gather(sizes, max_n, group=group)
    maxn = int(max(s.item() for s in sizes))
    padded = inp.new_zeros((maxn,) + tuple(inp.shape[1:]))
    padded[: inp.shape[0]] = inp
    gathered = [torch.zeros_like(padded) for _ in range(ws)]
    dist.all_gather(gathered, padded, group=group)
    # reconstruct: out receives, from each rank r, that rank's slice destined
    # to me (their in_splits are not known here, so exchange them too)
    splits_t = torch.tensor(in_splits, dtype=torch.long)
    all_splits = [torch.zeros_like(splits_t) for _ in range(ws)]
    dist.all_gather(all_splits, splits_t, group=group)
    chunks = []
    for r in range(ws):
        spl = all_splits[r].tolist()
        start = sum(spl[:rank])
        chunks.append(gathered[r][start:start + spl[rank]])
    result = torch.cat(chunks, dim=0)
    out.copy_(result)


class ExpertDispatch(torch.autograd.Function):
    """Differentiable all-to-all (backward = reverse all-to-all)."""

    @staticmethod
    def forward(ctx, inp: torch.Tensor, out_splits: List[int],
                in_splits: List[int], group) -> torch.Tensor:
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        ctx.group = group
        out = inp.new_empty((sum(out_splits),) + tuple(inp.shape[1:]))
        _all_to_all_single(out, inp.contiguous(), out_splits, in_splits, group)
        return out

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        out = grad.new_empty((sum(ctx.in_splits),) + tuple(grad.shape[1:]))
        _all_to_all_single(out, grad.contiguous(), ctx.in_splits,
                           ctx.out_splits, ctx.group)
        return out, None, None, None


def expert_all_to_all(inp: torch.Tensor, out_splits: List[int],
                      in_splits: List[int], group=None) -> torch.Tensor:
    return ExpertDispatch.apply(inp, out_splits, in_splits, group)


def exchange_counts(counts: torch.Tensor,
                    group=None) -> torch.Tensor:
    """counts [ep_size] (tokens this rank sends to each rank) ->
    [ep_size] (tokens this rank receives from each rank)."""
    ws = dist.get_world_size(group)
    gathered = [torch.zeros_like(counts) for _ in range(ws)]
    dist.all_gather(gathered, counts, group=group)
    rank = dist.get_rank(group)
    return torch.stack([g[rank] for g in gathered])

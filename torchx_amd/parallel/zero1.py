"""ZeRO stage 1 over the flat buffers: reduce-scatter gradients to
per-rank shards, shard the fp32 master/moment state, all-gather the
updated bf16 params.

Versus FlatDDP + full-state FlatAdamW:
  * wire traffic per step: 1x grads (reduce-scatter) + 1x params
    (all-gather) instead of 2x grads (ring all-reduce) — and the
    all-gather payload is bf16 params, half the fp32-equivalent;
  * optimizer memory (fp32 master + m + v = 12 bytes/param) drops by
    world_size — on 8x MI355X a Llama-3-8B's optimizer state goes from
    ~96 GB to ~12 GB per GPU.

The flat layout makes the sharding trivial: every group buffer is a
multiple of 64 elements, world sizes divide 64, so shard r is the
contiguous slice [r*n/ws, (r+1)*n/ws) and NCCL's in-place all-gather
(input = output's own slice) applies directly. RCCL reduce-scatter uses
in-collective AVG; the gloo CPU-CI fallback all-reduces SUM and slices.

Opt-in (``bench.py --zero1``); the default data-parallel path stays
FlatDDP (backward-overlapped buckets).
"""

from __future__ import annotations

from typing import Dict, Optional, Set

import torch
import torch.distributed as dist

from torchx_amd import ops

from .flat import FlatParams


class FlatZeRO1:
    def __init__(
        self,
        flat: FlatParams,
        process_group: Optional[dist.ProcessGroup] = None,
        lr: float = 3e-4,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.1,
        local_groups: Optional[Set[str]] = None,
    ):
        self.flat = flat
        self.pg = process_group
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.local_groups = local_groups or set()
        self.step_count = 0

        self.enabled = dist.is_available() and dist.is_initialized() and (
            dist.get_world_size(process_group) > 1
        )
        self.ws = dist.get_world_size(process_group) if self.enabled else 1
        self.rank = dist.get_rank(process_group) if self.enabled else 0
        self.use_nccl = (
            self.enabled and dist.get_backend(process_group) == "nccl"
        )

        # per-group shard bounds + sharded fp32 state (+ a grad-shard
        # scratch for the reduce-scatter output)
        self.shard: Dict[str, tuple] = {}
        self.state: Dict[str, Dict[str, torch.Tensor]] = {}
        self._gscratch: Dict[str, torch.Tensor] = {}
        for g, p16, grad in flat.groups():
            n = p16.numel()
            if g in self.local_groups or not self.enabled:
                lo, hi = 0, n
            else:
                assert n % self.ws == 0, (g, n, self.ws)
                s = n // self.ws
                lo, hi = self.rank * s, (self.rank + 1) * s
            self.shard[g] = (lo, hi)
            self.state[g] = {
                "p32": p16[lo:hi].float(),
                "m": torch.zeros(hi - lo, dtype=torch.float32,
                                 device=p16.device),
                "v": torch.zeros(hi - lo, dtype=torch.float32,
                                 device=p16.device),
            }
            if self.enabled and g not in self.local_groups:
                self._gscratch[g] = torch.empty(
                    hi - lo, dtype=grad.dtype, device=grad.device)

        if self.enabled:
            for g, p16, _ in flat.groups():
                if g not in self.local_groups:
                    dist.broadcast(p16, src=0, group=self.pg)

    def zero_grad(self) -> None:
        self.flat.zero_grad()

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        for g, p16, grad in self.flat.groups():
            lo, hi = self.shard[g]
            st = self.state[g]
            if self.enabled and g not in self.local_groups:
                if self.use_nccl:
                    # RCCL: in-collective AVG reduce-scatter
                    dist.reduce_scatter_tensor(
                        self._gscratch[g], grad,
                        op=dist.ReduceOp.AVG, group=self.pg,
                    )
                    gshard = self._gscratch[g]
                else:
                    # gloo lacks reduce_scatter_tensor/AVG
                    dist.all_reduce(grad, op=dist.ReduceOp.SUM, group=self.pg)
                    grad.div_(self.ws)
                    gshard = grad[lo:hi]
            else:
                if self.enabled:  # EP-local group: match the global mean
                    grad.div_(self.ws)
                gshard = grad[lo:hi]

            wd = 0.0 if g == "no_decay" else self.weight_decay
            ops.adamw_step(
                st["p32"], p16[lo:hi], gshard, st["m"], st["v"],
                lr=self.lr, beta1=self.beta1, beta2=self.beta2,
                eps=self.eps, weight_decay=wd, step=self.step_count,
            )
            if self.enabled and g not in self.local_groups:
                if self.use_nccl:
                    # in-place: each rank's input is its own output slice
                    dist.all_gather_into_tensor(p16, p16[lo:hi],
                                                group=self.pg)
                else:
                    shards = list(p16.chunk(self.ws))
                    dist.all_gather(shards, p16[lo:hi].contiguous(),
                                    group=self.pg)

    # FlatDDP-interface compatibility: ZeRO-1 reduces at step(), so the
    # trainer's ddp.finish() slot is a no-op here
    def finish(self) -> None:
        pass

    def state_dict(self) -> Dict:
        return {
            "step": self.step_count,
            "lr": self.lr,
            "rank": self.rank,
            "world_size": self.ws,
            "local_groups": sorted(self.local_groups),
            "state": {g: dict(st) for g, st in self.state.items()},
        }

    @staticmethod
    def consolidate(shards) -> Dict:
        """Merge per-rank ZeRO-1 ``state_dict``s into one full
        FlatAdamW-compatible state dict (for world-size changes or
        single-process analysis). ``shards`` is the list of all ranks'
        dicts, any order. Local (EP-sharded) groups keep rank 0's copy —
        they are rank-local by construction."""
        shards = sorted(shards, key=lambda s: s["rank"])
        ws = shards[0]["world_size"]
        if len(shards) != ws or [s["rank"] for s in shards] != list(range(ws)):
            raise ValueError(
                f"need all {ws} rank shards exactly once, got ranks "
                f"{[s.get('rank') for s in shards]}"
            )
        local = set(shards[0].get("local_groups", []))
        state: Dict[str, Dict[str, torch.Tensor]] = {}
        for g in shards[0]["state"]:
            if g in local:
                state[g] = dict(shards[0]["state"][g])
            else:
                state[g] = {
                    k: torch.cat([s["state"][g][k] for s in shards])
                    for k in shards[0]["state"][g]
                }
        return {"step": shards[0]["step"], "lr": shards[0]["lr"],
                "state": state}

    def load_state_dict(self, sd: Dict) -> None:
        assert sd.get("world_size", 1) == self.ws, (
            "ZeRO-1 checkpoints are sharded; resume with the same world "
            "size or consolidate first"
        )
        self.step_count = sd["step"]
        self.lr = sd["lr"]
        for g, st in sd["state"].items():
            for k, v in st.items():
                self.state[g][k].copy_(v)

"""Distributed execution context (SPMD over RCCL).

One process per GPU; every rank holds a shard of each table and runs the
same physical plan. Exchange points (shuffle/broadcast/merge) are RCCL
collectives — the re-imagining of the reference's Arrow Flight shuffle
(ref: crates/sail-execution/src/stream/, SURVEY §5.8 mapping).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class DistContext:
    dist: object  # torch.distributed module
    rank: int
    world: int
    device: str

    @property
    def is_distributed(self) -> bool:
        return self.world > 1

    def barrier(self):
        self.dist.barrier()

    def comm_device(self):
        """Collectives must run on the backend's device: nccl (RCCL) only
        accepts CUDA tensors, gloo only CPU — a CPU tensor on nccl is a
        hard error on the 8-GPU path that CPU gloo tests can't catch."""
        return self.device if self.dist.get_backend() == "nccl" else "cpu"

    def consensus_sum(self, value: int) -> int:
        """all_reduce(SUM) a host integer on the correct device."""
        t = torch.tensor([int(value)], dtype=torch.int64,
                         device=self.comm_device())
        self.dist.all_reduce(t, op=self.dist.ReduceOp.SUM)
        return int(t.item())

    def consensus_minmax(self, lo: int, hi: int):
        """all_reduce global (min, max) of per-rank ints (MAX of (-lo, hi))."""
        t = torch.tensor([-int(lo), int(hi)], dtype=torch.int64,
                         device=self.comm_device())
        self.dist.all_reduce(t, op=self.dist.ReduceOp.MAX)
        return -int(t[0].item()), int(t[1].item())

    def consensus_max(self, value: int) -> int:
        t = torch.tensor([int(value)], dtype=torch.int64,
                         device=self.comm_device())
        self.dist.all_reduce(t, op=self.dist.ReduceOp.MAX)
        return int(t.item())

    def all_reduce_sum_(self, t: torch.Tensor):
        self.dist.all_reduce(t, op=self.dist.ReduceOp.SUM)
        return t

    def all_gather_tensors(self, t: torch.Tensor):
        """All-gather variable-length 1-D tensors; returns list per rank."""
        n = torch.tensor([t.shape[0]], dtype=torch.int64, device=t.device)
        sizes = [torch.zeros_like(n) for _ in range(self.world)]
        self.dist.all_gather(sizes, n)
        sizes = [int(s.item()) for s in sizes]
        mx = max(sizes)
        pad = torch.zeros(mx, dtype=t.dtype, device=t.device)
        if t.shape[0]:
            pad[: t.shape[0]] = t
        outs = [torch.zeros_like(pad) for _ in range(self.world)]
        self.dist.all_gather(outs, pad)
        return [o[:s] for o, s in zip(outs, sizes)]

    def all_to_all_single(self, out: torch.Tensor, inp: torch.Tensor,
                          out_splits, in_splits):
        self.dist.all_to_all_single(out, inp, out_splits, in_splits)
        return out

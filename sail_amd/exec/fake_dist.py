"""Two-thread fake communicator with RCCL semantics on ONE device.

RCCL (like NCCL) refuses two ranks on one GPU, so the device-tensor
collective paths cannot be exercised on a 1-GPU box with real process
groups. This fake runs `world` ranks as THREADS in one process: each
collective checks the preconditions RCCL enforces (device tensors,
contiguity, matching dtypes, exact split-size sums) and then performs the
data movement with device copies. The exchange code runs the REAL
backend=="nccl" branches (get_backend() reports "nccl"), so dtype/stream/
size-math bugs surface before the driver's first 8-GPU run.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

import torch


class FakeRcclGroup:
    """Shared state for `world` thread-ranks."""

    def __init__(self, world: int, device: str, strict_cuda: bool = True):
        self.world = world
        self.device = device
        self.strict_cuda = strict_cuda
        self._barrier = threading.Barrier(world)
        self._slots: Dict[str, list] = {}
        self._lock = threading.Lock()
        self._seq = 0

    def rank_view(self, rank: int) -> "FakeRcclDist":
        return FakeRcclDist(self, rank)

    def _exchange(self, rank: int, value):
        """Deposit this rank's value, wait for all, return the full list."""
        with self._lock:
            key = str(self._seq // self.world)
            slot = self._slots.setdefault(key, [None] * self.world)
            slot[rank] = value
            self._seq += 1
        self._barrier.wait()
        out = self._slots[key]
        self._barrier.wait()
        if rank == 0:
            self._slots.pop(key, None)
        return out


class _ReduceOps:
    SUM = "sum"
    MAX = "max"
    MIN = "min"


class FakeRcclDist:
    """The torch.distributed surface DistContext uses, RCCL-checked."""

    ReduceOp = _ReduceOps

    def __init__(self, group: FakeRcclGroup, rank: int):
        self.group = group
        self.rank = rank

    # -- introspection ------------------------------------------------------
    def get_backend(self) -> str:
        return "nccl"

    def get_rank(self) -> int:
        return self.rank

    def get_world_size(self) -> int:
        return self.group.world

    def _check(self, t: torch.Tensor, what: str):
        if self.group.strict_cuda:
            assert t.is_cuda, f"RCCL {what}: tensor must be on the device"
        assert t.is_contiguous(), f"RCCL {what}: tensor must be contiguous"

    # -- collectives --------------------------------------------------------
    def barrier(self):
        self.group._barrier.wait()

    def all_reduce(self, tensor: torch.Tensor, op=_ReduceOps.SUM):
        self._check(tensor, "all_reduce")
        vals = self.group._exchange(self.rank, tensor.clone())
        dts = {v.dtype for v in vals}
        assert len(dts) == 1, f"RCCL all_reduce: dtype mismatch {dts}"
        shp = {tuple(v.shape) for v in vals}
        assert len(shp) == 1, f"RCCL all_reduce: shape mismatch {shp}"
        acc = vals[0].clone()
        for v in vals[1:]:
            if op == _ReduceOps.SUM or op is None:
                acc = acc + v
            elif op == _ReduceOps.MAX:
                acc = torch.maximum(acc, v)
            elif op == _ReduceOps.MIN:
                acc = torch.minimum(acc, v)
            else:
                raise AssertionError(f"unsupported op {op}")
        tensor.copy_(acc)

    def all_gather(self, out_list: List[torch.Tensor], tensor: torch.Tensor):
        self._check(tensor, "all_gather")
        vals = self.group._exchange(self.rank, tensor.clone())
        assert len(out_list) == self.group.world
        for dst, src in zip(out_list, vals):
            assert dst.shape == src.shape, \
                f"RCCL all_gather: buffer shape {dst.shape} != {src.shape}"
            assert dst.dtype == src.dtype
            dst.copy_(src)

    def all_to_all_single(self, output: torch.Tensor, input_: torch.Tensor,
                          output_split_sizes: Optional[List[int]] = None,
                          input_split_sizes: Optional[List[int]] = None):
        self._check(output, "all_to_all_single(out)")
        self._check(input_, "all_to_all_single(in)")
        w = self.group.world
        if input_split_sizes is None:
            assert input_.numel() % w == 0, \
                "RCCL all_to_all_single: equal-split input not divisible"
            input_split_sizes = [input_.numel() // w] * w
        if output_split_sizes is None:
            assert output.numel() % w == 0
            output_split_sizes = [output.numel() // w] * w
        assert sum(input_split_sizes) == input_.numel(), \
            "RCCL all_to_all_single: input splits don't sum to numel"
        assert sum(output_split_sizes) == output.numel(), \
            "RCCL all_to_all_single: output splits don't sum to numel"
        # slice per destination, exchange, reassemble per source
        pieces = []
        at = 0
        for n in input_split_sizes:
            pieces.append(input_[at:at + n].clone())
            at += n
        all_pieces = self.group._exchange(self.rank, pieces)
        at = 0
        for src in range(w):
            piece = all_pieces[src][self.rank]
            n = output_split_sizes[src]
            assert piece.numel() == n, \
                (f"RCCL all_to_all_single: rank {self.rank} expected {n} "
                 f"from rank {src}, got {piece.numel()}")
            assert piece.dtype == output.dtype, \
                "RCCL all_to_all_single: dtype mismatch"
            output[at:at + n].copy_(piece)
            at += n


def run_world(world: int, device: str, fn, strict_cuda: bool = True):
    """Run fn(rank, dist) on `world` threads with a shared fake group;
    re-raises the first failure."""
    group = FakeRcclGroup(world, device, strict_cuda=strict_cuda)
    errors: List[BaseException] = []

    def runner(r):
        try:
            fn(r, group.rank_view(r))
        except BaseException as e:  # noqa: BLE001 — surfaced to caller
            errors.append(e)
            # release peers stuck on the barrier
            group._barrier.abort()

    threads = [threading.Thread(target=runner, args=(r,), daemon=True)
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=600)
    if errors:
        raise errors[0]

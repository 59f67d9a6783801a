"""Positional column batch used by the executor.

Operators pass `Chunk`s (ordered columns + names); names may repeat after
joins, so lookups are ordinal (BoundRef.index), never by name.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .column import Column, StringColumn, Table


@dataclass
class Chunk:
    columns: List[Column] = field(default_factory=list)
    names: List[str] = field(default_factory=list)
    #: distribution across ranks in SPMD mode: "sharded" (each rank holds a
    #: horizontal slice) or "replicated" (identical on every rank). Ignored
    #: in single-process execution.
    partitioning: str = "replicated"
    #: row-count override for sparse chunks (lazy Project∘Filter fusion may
    #: leave unreferenced column slots as None)
    forced_rows: "Optional[int]" = None

    @property
    def num_rows(self) -> int:
        if self.forced_rows is not None:
            return self.forced_rows
        for c in self.columns:
            if c is not None:
                return len(c)
        return 0

    @property
    def device(self) -> torch.device:
        for c in self.columns:
            if c is not None:
                return c.device
        return torch.device("cpu")

    def gather(self, indices: torch.Tensor) -> "Chunk":
        return Chunk([c.gather(indices) for c in self.columns], list(self.names),
                     self.partitioning)

    def filter_mask(self, mask: torch.Tensor) -> "Chunk":
        idx = torch.nonzero(mask, as_tuple=False).squeeze(1)
        return self.gather(idx)

    def slice(self, start: int, length: int) -> "Chunk":
        return Chunk([c.slice(start, length) for c in self.columns], list(self.names),
                     self.partitioning)

    def to_table(self) -> Table:
        cols = {}
        for n, c in zip(self.names, self.columns):
            name, k = n, 1
            while name in cols:
                k += 1
                name = f"{n}_{k}"
            cols[name] = c
        return Table(cols)

    def to(self, device) -> "Chunk":
        return Chunk([c.to(device) for c in self.columns], list(self.names),
                     self.partitioning)

    @staticmethod
    def from_table(t: Table) -> "Chunk":
        return Chunk(list(t.columns.values()), list(t.columns.keys()))

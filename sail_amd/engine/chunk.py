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

    @property
    def num_rows(self) -> int:
        return len(self.columns[0]) if self.columns else 0

    @property
    def device(self) -> torch.device:
        return self.columns[0].device if self.columns else torch.device("cpu")

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

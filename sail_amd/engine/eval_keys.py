"""Literal-side key computation matching joins.raw_string_key schemes."""
import torch

from .joins import _FNV_OFFSET, _FNV_PRIME, _M64, _MIX, fnv1a_hash_py


def pack7_py(b: bytes) -> int:
    k = len(b)
    for i in range(7):
        k = k * 257 + (b[i] if i < len(b) else 0)
    return k


def literal_key(s: str, force_hash: bool) -> int:
    b = s.encode("utf-8")
    if not force_hash and len(b) <= 7:
        return pack7_py(b)
    return fnv1a_hash_py(b)


def literal_keys_like(values, force_hash: bool, device) -> torch.Tensor:
    return torch.tensor([literal_key(v, force_hash) for v in values],
                        dtype=torch.int64, device=device)

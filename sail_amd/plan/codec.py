"""Plan codec: spec Plan/Expr trees <-> JSON.

The reference serializes physical plans to ship them driver->worker
(ref: crates/sail-execution plan codec). This engine's SPMD
design re-resolves plans per rank, so the codec's jobs here are plan
persistence/transport over the Connect boundary, gold snapshots, and
debugging — a faithful round-trip for every Plan/Expr dataclass in
plan/spec.py. `ChunkSource` (pre-materialized device data) is explicitly
not serializable.
"""
from __future__ import annotations

import dataclasses
import json
from typing import Any

from ..engine import types as T
from . import spec as S


class CodecError(ValueError):
    pass


# -- data types -------------------------------------------------------------

def type_to_obj(t) -> Any:
    if t is None:
        return None
    if isinstance(t, T.ArrayType):
        return {"t": "array", "e": type_to_obj(t.element)}
    if isinstance(t, T.MapType):
        return {"t": "map", "k": type_to_obj(t.key),
                "v": type_to_obj(t.value)}
    if isinstance(t, T.StructType):
        return {"t": "struct",
                "fields": [[f.name, type_to_obj(f.dtype)]
                           for f in t.fields]}
    return {"t": T.type_name(t)}


def type_from_obj(o) -> Any:
    if o is None:
        return None
    k = o["t"]
    if k == "array":
        return T.ArrayType(type_from_obj(o["e"]))
    if k == "map":
        return T.MapType(type_from_obj(o["k"]), type_from_obj(o["v"]))
    if k == "struct":
        return T.StructType(tuple(
            T.StructField(n, type_from_obj(ft)) for n, ft in o["fields"]))
    return T.type_from_name(k)


# -- literal values ---------------------------------------------------------

def _val_to_obj(v):
    import datetime as _dt
    from decimal import Decimal

    if isinstance(v, (bytes, bytearray)):
        import base64

        return {"_b": base64.b64encode(bytes(v)).decode()}
    if isinstance(v, Decimal):
        return {"_dec": str(v)}
    if isinstance(v, _dt.datetime):
        return {"_ts": v.isoformat()}
    if isinstance(v, _dt.date):
        return {"_d": v.isoformat()}
    if isinstance(v, tuple):
        return {"_tup": [_val_to_obj(x) for x in v]}
    if isinstance(v, list):
        return {"_list": [_val_to_obj(x) for x in v]}
    if isinstance(v, dict):
        return {"_dict": [[_val_to_obj(k), _val_to_obj(x)]
                          for k, x in v.items()]}
    return v


def _val_from_obj(o):
    import datetime as _dt
    from decimal import Decimal

    if isinstance(o, dict):
        if "_b" in o:
            import base64

            return base64.b64decode(o["_b"])
        if "_dec" in o:
            return Decimal(o["_dec"])
        if "_ts" in o:
            return _dt.datetime.fromisoformat(o["_ts"])
        if "_d" in o:
            return _dt.date.fromisoformat(o["_d"])
        if "_tup" in o:
            return tuple(_val_from_obj(x) for x in o["_tup"])
        if "_list" in o:
            return [_val_from_obj(x) for x in o["_list"]]
        if "_dict" in o:
            return {_val_from_obj(k): _val_from_obj(x)
                    for k, x in o["_dict"]}
    return o


# -- trees ------------------------------------------------------------------

_NODE_CLASSES = {}
for _nm in dir(S):
    _c = getattr(S, _nm)
    if isinstance(_c, type) and issubclass(_c, (S.Plan, S.Expr)) \
            and dataclasses.is_dataclass(_c):
        _NODE_CLASSES[_nm] = _c


def to_obj(x) -> Any:
    if x is None or isinstance(x, (bool, int, float, str)):
        return x
    if isinstance(x, S.ChunkSource):
        raise CodecError("ChunkSource (materialized data) is not "
                         "serializable")
    if isinstance(x, (S.Plan, S.Expr)):
        d = {"_k": type(x).__name__}
        for f in dataclasses.fields(x):
            v = getattr(x, f.name)
            if f.name in ("dtype",) or (f.name == "schema" and
                                        _is_schema(v)):
                d[f.name] = _schema_to_obj(v) if f.name == "schema" \
                    else type_to_obj(v)
            elif isinstance(x, S.Literal) and f.name == "value":
                d[f.name] = _val_to_obj(v)
            else:
                d[f.name] = to_obj(v)
        return d
    if isinstance(x, T.DataType):
        return {"_dt": type_to_obj(x)}
    if isinstance(x, (list, tuple)):
        return [to_obj(v) for v in x]
    if isinstance(x, dict):
        return {str(k): to_obj(v) for k, v in x.items()}
    raise CodecError(f"unserializable node {type(x).__name__}")


def _is_schema(v):
    return isinstance(v, list) and v and isinstance(v[0], (tuple, list)) \
        and len(v[0]) == 2 and isinstance(v[0][1], T.DataType)


def _schema_to_obj(v):
    if v is None:
        return None
    if not _is_schema(v):
        return to_obj(v)
    return {"_schema": [[n, type_to_obj(t)] for n, t in v]}


def from_obj(o) -> Any:
    if o is None or isinstance(o, (bool, int, float, str)):
        return o
    if isinstance(o, list):
        return [from_obj(v) for v in o]
    if isinstance(o, dict):
        if "_k" in o:
            cls = _NODE_CLASSES.get(o["_k"])
            if cls is None:
                raise CodecError(f"unknown node kind {o['_k']}")
            kwargs = {}
            for f in dataclasses.fields(cls):
                if f.name not in o:
                    continue
                v = o[f.name]
                if f.name == "dtype":
                    kwargs[f.name] = type_from_obj(v)
                elif f.name == "schema" and isinstance(v, dict) \
                        and "_schema" in v:
                    kwargs[f.name] = [(n, type_from_obj(t))
                                      for n, t in v["_schema"]]
                elif cls is S.Literal and f.name == "value":
                    kwargs[f.name] = _val_from_obj(v)
                else:
                    kwargs[f.name] = from_obj(v)
            return cls(**kwargs)
        if "_dt" in o:
            return type_from_obj(o["_dt"])
        if "_schema" in o:
            return [(n, type_from_obj(t)) for n, t in o["_schema"]]
        return {k: from_obj(v) for k, v in o.items()}
    return o


def plan_to_json(plan: S.Plan, indent=None) -> str:
    return json.dumps(to_obj(plan), indent=indent)


def plan_from_json(text: str) -> S.Plan:
    return from_obj(json.loads(text))

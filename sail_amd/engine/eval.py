"""Column-wise expression evaluation.

This is the engine's *reference* implementation of every expression and
scalar function: pure torch ops over whole columns, so it runs identically on
CPU (tests) and on ROCm (fallback path). The hot GPU path replaces entire
filter/project trees with one fused ExprVM kernel (ops/csrc/exprvm.hip); its
numerics are validated against this module.

Null semantics follow Spark/SQL three-valued logic
(ref: crates/sail-function/src/scalar/* for per-function behavior).
"""
from __future__ import annotations

import datetime as _dt
import math
import re
from typing import List, Optional, Union

import torch

from ..plan import spec as S
from . import types as T
from .chunk import Chunk
from .column import Column, StringColumn, _pack_strings


class EvalError(Exception):
    pass


# ---------------------------------------------------------------------------
# Scalar wrapper: literal values flow through evaluation unexpanded
# ---------------------------------------------------------------------------

class Scalar:
    __slots__ = ("value", "dtype")

    def __init__(self, value, dtype: T.DataType):
        self.value = value
        self.dtype = dtype

    @property
    def is_null(self):
        return self.value is None


Val = Union[Column, Scalar]


def broadcast(v: Val, n: int, device) -> Column:
    """Materialize a Scalar to a full column."""
    if isinstance(v, Column):
        return v
    dt = v.dtype
    if isinstance(v.value, tuple) and len(v.value) == 3 \
            and v.value[0] == "__interval__":
        # bare interval literal (SELECT make_dt_interval(...)): render in
        # Spark's display form; arithmetic consumes the tuple upstream
        _, months, micros = v.value
        return broadcast(Scalar(_format_interval(months, micros), T.STRING),
                         n, device)
    if v.is_null:
        storage = dt.storage or torch.int64
        data = torch.zeros(n, dtype=storage if not isinstance(dt, T.StringType) else torch.int64, device=device)
        if isinstance(dt, T.StringType):
            offs = torch.zeros(n + 1, dtype=torch.int64, device=device)
            return StringColumn(offs, torch.zeros(0, dtype=torch.uint8, device=device),
                                torch.zeros(n, dtype=torch.uint8, device=device))
        return Column(dt, data, torch.zeros(n, dtype=torch.uint8, device=device))
    if isinstance(dt, T.StringType):
        offs, byts = _pack_strings([v.value], device)
        codes = torch.zeros(n, dtype=torch.int32, device=device)
        return StringColumn(offs, byts, None, codes, dtype=dt)
    val = v.value
    if isinstance(dt, T.DecimalType):
        val = _to_scaled(val, dt.scale)
    elif isinstance(dt, T.DateType):
        if isinstance(val, str):
            val = _date_str_to_days(val)
        elif isinstance(val, _dt.date):
            val = (val - _dt.date(1970, 1, 1)).days
    elif isinstance(dt, T.TimeType) and isinstance(val, _dt.time):
        val = ((val.hour * 60 + val.minute) * 60 + val.second) * 1_000_000 \
            + val.microsecond
    data = torch.full((n,), val, dtype=dt.storage, device=device)
    return Column(dt, data, None)


def _format_interval(months: int, micros: int) -> str:
    """Spark display form for a folded interval literal."""
    parts = []
    if months:
        y, m = divmod(abs(months), 12)
        sign = "-" if months < 0 else ""
        if y:
            parts.append(f"{sign}{y} years")
        if m:
            parts.append(f"{sign}{m} months")
    if micros or not months:
        sign = "-" if micros < 0 else ""
        us = abs(micros)
        d, us = divmod(us, 86_400_000_000)
        h, us = divmod(us, 3_600_000_000)
        mi, us = divmod(us, 60_000_000)
        s = us / 1_000_000
        if d:
            parts.append(f"{sign}{d} days")
        if h:
            parts.append(f"{sign}{h} hours")
        if mi:
            parts.append(f"{sign}{mi} minutes")
        if s or not parts:
            ss = f"{s:g}"
            parts.append(f"{sign}{ss} seconds")
    return "INTERVAL '" + " ".join(parts) + "'"


def _to_scaled(v, scale: int) -> int:
    if isinstance(v, int):
        return v * (10 ** scale)
    # round-half-up like Spark
    from decimal import Decimal, ROUND_HALF_UP

    return int(Decimal(str(v)).scaleb(scale).quantize(Decimal(1), rounding=ROUND_HALF_UP))


def _date_str_to_days(s: str) -> int:
    y, m, d = s.strip().split("-")[:3]
    return (_dt.date(int(y), int(m), int(d)) - _dt.date(1970, 1, 1)).days


# ---------------------------------------------------------------------------
# Evaluator
# ---------------------------------------------------------------------------

class Evaluator:
    """Evaluates bound expressions against a Chunk."""

    def __init__(self, ctx=None):
        self.ctx = ctx  # execution context (for subquery results, configs)
        self.subquery_values = {}  # id(ScalarSubquery) -> Scalar

    # -- public ------------------------------------------------------------
    def eval(self, e: S.Expr, chunk: Chunk) -> Val:
        m = getattr(self, "_e_" + type(e).__name__, None)
        if m is None:
            raise EvalError(f"cannot evaluate {type(e).__name__}")
        return m(e, chunk)

    def eval_col(self, e: S.Expr, chunk: Chunk) -> Column:
        return broadcast(self.eval(e, chunk), chunk.num_rows, chunk.device)

    def eval_mask(self, e: S.Expr, chunk: Chunk) -> torch.Tensor:
        """Evaluate a predicate to a boolean tensor (null -> False)."""
        v = self.eval(e, chunk)
        if isinstance(v, Scalar):
            bit = bool(v.value) if v.value is not None else False
            return torch.full((chunk.num_rows,), bit, dtype=torch.bool, device=chunk.device)
        mask = v.data
        if mask.dtype != torch.bool:
            mask = mask != 0
        if v.validity is not None:
            mask = mask & v.validity.to(torch.bool)
        return mask

    # -- leaves ------------------------------------------------------------
    def _e_Literal(self, e: S.Literal, chunk: Chunk) -> Scalar:
        return Scalar(e.value, e.dtype or T.NULL)

    def _e_BoundRef(self, e: S.BoundRef, chunk: Chunk) -> Column:
        return chunk.columns[e.index]

    def _e_Alias(self, e: S.Alias, chunk: Chunk) -> Val:
        return self.eval(e.child, chunk)

    def _e_ScalarSubquery(self, e: S.ScalarSubquery, chunk: Chunk) -> Scalar:
        v = self.subquery_values.get(id(e))
        if v is None:
            if self.ctx is None:
                raise EvalError("scalar subquery not pre-executed")
            v = self.ctx.execute_scalar_subquery(e)
            self.subquery_values[id(e)] = v
        return v

    # -- operators ---------------------------------------------------------
    def _e_BinaryOp(self, e: S.BinaryOp, chunk: Chunk) -> Val:
        if e.op in ("and", "or"):
            return self._kleene(e, chunk)
        l = self.eval(e.left, chunk)
        r = self.eval(e.right, chunk)
        if isinstance(l, Scalar) and isinstance(r, Scalar):
            return _scalar_binop(e.op, l, r, e.dtype)
        n, dev = chunk.num_rows, chunk.device
        if e.op == "<=>":
            lc, rc = broadcast(l, n, dev), broadcast(r, n, dev)
            lv, rv = lc.valid_mask(), rc.valid_mask()
            eq = _cmp_data(lc, rc, "=")
            out = (lv & rv & eq) | (~lv & ~rv)
            return Column(T.BOOL, out, None)
        if e.op in ("=", "!=", "<", "<=", ">", ">="):
            # column-vs-scalar fast path: tensor-op-pyscalar, no broadcast
            for col, sc, op in ((l, r, e.op), (r, l, _flip_cmp(e.op))):
                if isinstance(col, Column) and isinstance(sc, Scalar) \
                        and not isinstance(col, StringColumn) and not sc.is_null \
                        and not isinstance(sc.value, str):
                    v = sc.value
                    if isinstance(v, _dt.date):
                        v = (v - _dt.date(1970, 1, 1)).days
                    if isinstance(col.dtype, T.DecimalType):
                        # rescale the scalar exactly to the column's scale
                        v = _to_scaled(v, col.dtype.scale)
                    elif isinstance(v, float) and not col.data.dtype.is_floating_point:
                        # int column vs fractional literal: fall through to
                        # the generic coerced path
                        if v != int(v):
                            break
                        v = int(v)
                    data = _cmp_tensor_scalar(col.data, v, op)
                    return Column(T.BOOL, data, col.validity)
            lc, rc = broadcast(l, n, dev), broadcast(r, n, dev)
            data = _cmp_data(lc, rc, e.op)
            return Column(T.BOOL, data, _merge_validity(lc, rc))
        # arithmetic
        lc, rc = broadcast(l, n, dev), broadcast(r, n, dev)
        return _arith(e.op, lc, rc, e.dtype)

    def _eval_udf(self, e: S.Func, chunk: Chunk) -> Val:
        """Host Python UDF: device columns -> host lists -> fn -> column."""
        sess = self.ctx.session if self.ctx is not None else None
        info = sess.udfs.get(e.name.lower()) if sess is not None else None
        if info is None:
            raise EvalError(f"UDF {e.name} not registered")
        fn, rtype = info[0], info[1]
        vectorized = info[2] if len(info) > 2 else False
        from .column import Column as _C, StringColumn as _S

        if vectorized:
            # pandas/Arrow-batched UDF: one call over whole numpy arrays,
            # the reference's vectorized PySpark UDF path
            # (ref: crates/sail-python-udf pandas_udf) minus the IPC hop —
            # the arrays come straight off the device tensors.
            import numpy as np

            args = []
            for a in e.args:
                c = self.eval_col(a, chunk)
                if isinstance(c, _S):
                    args.append(np.array(c.to_pylist(), dtype=object))
                else:
                    args.append(c.data.cpu().numpy())
            out = fn(*args) if args else fn(chunk.num_rows)
            if isinstance(rtype, T.StringType):
                return _S.from_pylist(list(out), device=chunk.device)
            arr = np.asarray(out)
            data = torch.from_numpy(np.ascontiguousarray(arr))
            if rtype.storage is not None and data.dtype != rtype.storage:
                data = data.to(rtype.storage)
            return _C(rtype, data.to(chunk.device))
        cols = [self.eval_col(a, chunk).to_pylist() for a in e.args]
        out = [fn(*vals) for vals in zip(*cols)] if cols else [fn() for _ in range(chunk.num_rows)]
        if isinstance(rtype, T.StringType):
            return _S.from_pylist(out, device=chunk.device)
        return _C.from_values(out, rtype, device=chunk.device)

    def _kleene(self, e: S.BinaryOp, chunk: Chunk) -> Val:
        l = self.eval(e.left, chunk)
        r = self.eval(e.right, chunk)
        n, dev = chunk.num_rows, chunk.device
        lc, rc = broadcast(l, n, dev), broadcast(r, n, dev)
        if lc.validity is None and rc.validity is None:
            ld = lc.data if lc.data.dtype == torch.bool else lc.data != 0
            rd = rc.data if rc.data.dtype == torch.bool else rc.data != 0
            return Column(T.BOOL, ld & rd if e.op == "and" else ld | rd, None)
        lv, rv = lc.valid_mask(), rc.valid_mask()
        ld = lc.data.to(torch.bool)
        rd = rc.data.to(torch.bool)
        if e.op == "and":
            data = ld & rd
            # null unless (either false) or (both valid)
            valid = (lv & rv) | (lv & ~ld) | (rv & ~rd)
        else:
            data = (ld & lv) | (rd & rv)
            valid = (lv & rv) | (lv & ld) | (rv & rd)
        if bool(valid.all()):
            return Column(T.BOOL, data, None)
        return Column(T.BOOL, data, valid.to(torch.uint8))

    def _e_UnaryOp(self, e: S.UnaryOp, chunk: Chunk) -> Val:
        v = self.eval(e.child, chunk)
        if e.op == "not":
            if isinstance(v, Scalar):
                return Scalar(None if v.value is None else (not bool(v.value)), T.BOOL)
            data = ~v.data.to(torch.bool)
            return Column(T.BOOL, data, v.validity)
        if e.op == "isnull":
            if isinstance(v, Scalar):
                return Scalar(v.value is None, T.BOOL)
            return Column(T.BOOL, ~v.valid_mask(), None)
        if e.op == "isnotnull":
            if isinstance(v, Scalar):
                return Scalar(v.value is not None, T.BOOL)
            return Column(T.BOOL, v.valid_mask(), None)
        if e.op == "neg":
            if isinstance(v, Scalar):
                return Scalar(None if v.value is None else -v.value, v.dtype)
            return Column(v.dtype, -v.data, v.validity)
        raise EvalError(f"unknown unary {e.op}")

    def _e_Cast(self, e: S.Cast, chunk: Chunk) -> Val:
        v = self.eval(e.child, chunk)
        return cast_value(v, e.to, chunk, try_=e.try_)

    def _e_CaseWhen(self, e: S.CaseWhen, chunk: Chunk) -> Val:
        n, dev = chunk.num_rows, chunk.device
        if isinstance(e.dtype, T.StringType):
            return self._case_when_string(e, chunk)
        result = broadcast(self.eval(e.else_, chunk) if e.else_ is not None
                           else Scalar(None, e.dtype), n, dev)
        result_data = result.data.clone()
        valid = result.valid_mask().clone()
        decided = torch.zeros(n, dtype=torch.bool, device=dev)
        for cond, val in e.branches:
            cmask = self.eval_mask(cond, chunk) & ~decided
            vcol = broadcast(self.eval(val, chunk), n, dev)
            vdata = vcol.data
            if vdata.dtype != result_data.dtype:
                vdata = vdata.to(result_data.dtype)
            result_data = torch.where(cmask, vdata, result_data)
            valid = torch.where(cmask, vcol.valid_mask(), valid)
            decided |= cmask
        return Column(e.dtype, result_data, None if bool(valid.all()) else valid.to(torch.uint8))

    def _case_when_string(self, e: S.CaseWhen, chunk: Chunk) -> Val:
        """String-valued CASE: when all branch values are scalar strings the
        result is dictionary-encoded (selector codes over the branch values);
        otherwise assemble on host (reference path)."""
        n, dev = chunk.num_rows, chunk.device
        vals = [self.eval(v, chunk) for _, v in e.branches]
        elsev = self.eval(e.else_, chunk) if e.else_ is not None else Scalar(None, e.dtype)
        if all(isinstance(v, Scalar) for v in vals) and isinstance(elsev, Scalar):
            branch_strs = [v.value for v in vals] + [elsev.value]
            # dedup dictionary
            uniq = sorted({s for s in branch_strs if s is not None})
            idx = {s: i for i, s in enumerate(uniq)}
            codes = torch.full((n,), -1 if elsev.value is None else idx[elsev.value],
                               dtype=torch.int32, device=dev)
            decided = torch.zeros(n, dtype=torch.bool, device=dev)
            for (cond, _), v in zip(e.branches, vals):
                cmask = self.eval_mask(cond, chunk) & ~decided
                code = -1 if v.value is None else idx[v.value]
                codes = torch.where(cmask, torch.full_like(codes, code), codes)
                decided |= cmask
            offs, byts = _pack_strings(uniq, dev)
            validity = None
            if bool((codes < 0).any()):
                validity = (codes >= 0).to(torch.uint8)
            return StringColumn(offs, byts, validity, codes)
        # host fallback
        out: List[Optional[str]] = [None] * n
        decided = torch.zeros(n, dtype=torch.bool, device=dev)
        branch_lists = [broadcast(v, n, dev).to_pylist() for v in vals]
        else_list = broadcast(elsev, n, dev).to_pylist()
        masks = []
        for cond, _ in e.branches:
            m = self.eval_mask(cond, chunk) & ~decided
            decided |= m
            masks.append(m.cpu().tolist())
        for i in range(n):
            val = else_list[i]
            for bi, m in enumerate(masks):
                if m[i]:
                    val = branch_lists[bi][i]
                    break
            out[i] = val
        return StringColumn.from_pylist(out, device=dev)

    def _e_Between(self, e: S.Between, chunk: Chunk) -> Val:
        lo = S.BinaryOp(">=", e.child, e.low, T.BOOL)
        hi = S.BinaryOp("<=", e.child, e.high, T.BOOL)
        _retype_cmp(lo)
        _retype_cmp(hi)
        combined = S.BinaryOp("and", lo, hi, T.BOOL)
        v = self.eval(combined, chunk)
        return _negate(v) if e.negated else v

    def _e_InList(self, e: S.InList, chunk: Chunk) -> Val:
        child = self.eval(e.child, chunk)
        n, dev = chunk.num_rows, chunk.device
        cc = broadcast(child, n, dev)
        vals = [self.eval(x, chunk) for x in e.values]
        if isinstance(cc, StringColumn):
            lits = [v.value for v in vals if isinstance(v, Scalar)]
            mask = _string_isin(cc, lits)
        else:
            targets = []
            for v in vals:
                if not isinstance(v, Scalar):
                    # general IN: OR-chain of equality comparisons with SQL
                    # 3VL (true if any match; null if no match but a null
                    # comparison exists)
                    any_true = None
                    any_null = None
                    for ve in e.values:
                        cmpe = S.BinaryOp("=", e.child, ve, T.BOOL)
                        cv = self.eval_col(cmpe, chunk)
                        m = cv.data.to(torch.bool) & cv.valid_mask()
                        nl = ~cv.valid_mask()
                        any_true = m if any_true is None else (any_true | m)
                        any_null = nl if any_null is None else (any_null | nl)
                    valid = any_true | ~any_null
                    data = ~any_true if e.negated else any_true
                    return Column(T.BOOL, data,
                                  None if bool(valid.all()) else valid.to(torch.uint8))
                tv = v.value
                if tv is None:
                    targets.append(None)
                    continue
                if isinstance(cc.dtype, T.DecimalType):
                    tv = _to_scaled(tv, cc.dtype.scale)
                elif isinstance(cc.dtype, T.DateType) and isinstance(tv, str):
                    tv = _date_str_to_days(tv)
                targets.append(tv)
            has_null = any(t is None for t in targets)
            nn = [t for t in targets if t is not None]
            tt = torch.tensor(nn, dtype=cc.data.dtype, device=dev) if nn \
                else torch.zeros(0, dtype=cc.data.dtype, device=dev)
            mask = torch.isin(cc.data, tt)
            if has_null:
                # SQL 3VL: a NULL in the list makes non-matches NULL
                valid = cc.valid_mask() & mask
                data = ~mask if e.negated else mask
                return Column(T.BOOL, data,
                              None if bool(valid.all())
                              else valid.to(torch.uint8))
        if e.negated:
            mask = ~mask
        return Column(T.BOOL, mask, cc.validity)

    def _e_Like(self, e: S.Like, chunk: Chunk) -> Val:
        child = self.eval(e.child, chunk)
        pat = self.eval(e.pattern, chunk)
        if not isinstance(pat, Scalar):
            raise EvalError("LIKE pattern must be a literal")
        n, dev = chunk.num_rows, chunk.device
        cc = broadcast(child, n, dev)
        if not isinstance(cc, StringColumn):
            raise EvalError("LIKE on non-string")
        mask = string_like(cc, pat.value, case_insensitive=e.case_insensitive, is_regex=e.is_regex)
        if e.negated:
            mask = ~mask
        return Column(T.BOOL, mask, cc.validity)

    def _e_Func(self, e: S.Func, chunk: Chunk) -> Val:
        from .functions_impl import dispatch_function

        sess = getattr(self.ctx, "session", None) if self.ctx is not None else None
        if sess is not None and getattr(sess, "udfs", None) and e.name.lower() in sess.udfs:
            return self._eval_udf(e, chunk)
        if any(isinstance(a, S.Lambda) for a in e.args):
            from .arrays import eval_hof

            return eval_hof(self, e, chunk)
        args = [self.eval(a, chunk) for a in e.args]
        if args and all(isinstance(a, Scalar) for a in args) \
                and e.name not in ("rand", "randn", "uuid", "monotonically_increasing_id", "random", "uniform", "randstr") \
                and not isinstance(e.dtype, (T.ArrayType, T.MapType, T.StructType)) \
                and not any(isinstance(a.dtype, (T.ArrayType, T.MapType, T.StructType))
                            for a in args):
            # constant folding: evaluate once on a 1-row chunk
            one = Chunk([], [], chunk.partitioning)
            one.forced_rows = 1
            out = dispatch_function(e.name, args, e.dtype, one, self)
            if isinstance(out, Scalar):
                return out
            if isinstance(out.dtype, (T.ArrayType, T.MapType, T.StructType)):
                # nested results (e.g. from_protobuf on literals) can't be
                # scalarized: replicate the single row to the chunk length
                n = chunk.num_rows or 1
                rep = out.gather(torch.zeros(n, dtype=torch.int64,
                                             device=out.device))
                return rep.to(chunk.device)
            vals = out.to_pylist()
            v = vals[0] if vals else None
            if isinstance(e.dtype, T.DecimalType) and v is not None:
                # to_pylist already unscaled; keep as float for rebroadcast
                return Scalar(v, e.dtype)
            if isinstance(out.dtype, (T.GeometryType, T.GeographyType)):
                # st_setsrid refines the SRID at eval time; the resolver
                # type only knows the default
                return Scalar(v, out.dtype)
            return Scalar(v, e.dtype or out.dtype)
        return dispatch_function(e.name, args, e.dtype, chunk, self)


def _negate(v: Val) -> Val:
    if isinstance(v, Scalar):
        return Scalar(None if v.value is None else not bool(v.value), T.BOOL)
    return Column(T.BOOL, ~v.data.to(torch.bool), v.validity)


def _retype_cmp(e: S.BinaryOp):
    """Re-run comparison operand coercion for synthesized BinaryOps."""
    from ..plan.resolver import _coerce_pair

    if e.left.dtype is not None and e.right.dtype is not None and e.left.dtype != e.right.dtype:
        _coerce_pair(e)


# ---------------------------------------------------------------------------
# kernels (torch fallback implementations)
# ---------------------------------------------------------------------------

def _flip_cmp(op: str) -> str:
    return {"=": "=", "!=": "!=", "<": ">", "<=": ">=", ">": "<", ">=": "<="}[op]


def _cmp_tensor_scalar(x: torch.Tensor, v, op: str) -> torch.Tensor:
    if op == "=":
        return x == v
    if op == "!=":
        return x != v
    if op == "<":
        return x < v
    if op == "<=":
        return x <= v
    if op == ">":
        return x > v
    return x >= v


def _merge_validity(a: Column, b: Column) -> Optional[torch.Tensor]:
    if a.validity is None and b.validity is None:
        return None
    return (a.valid_mask() & b.valid_mask()).to(torch.uint8)


def _cmp_data(a: Column, b: Column, op: str) -> torch.Tensor:
    if isinstance(a, StringColumn) or isinstance(b, StringColumn):
        return _string_cmp(a, b, op)
    x, y = a.data, b.data
    if x.dtype != y.dtype:
        ct = torch.promote_types(x.dtype, y.dtype)
        x, y = x.to(ct), y.to(ct)
    if op == "=":
        return x == y
    if op == "!=":
        return x != y
    if op == "<":
        return x < y
    if op == "<=":
        return x <= y
    if op == ">":
        return x > y
    if op == ">=":
        return x >= y
    raise EvalError(op)


def _literal_string_keys(col: StringColumn, lits) -> torch.Tensor:
    """Literal keys consistent with joins.raw_string_key for `col`."""
    from .eval_keys import literal_keys_like

    lens = col.offsets[1:] - col.offsets[:-1]
    force_hash = (int(lens.max().item()) if len(col) else 0) > 7
    return literal_keys_like(list(lits), force_hash, col.device)


def _string_cmp(a: Column, b: Column, op: str) -> torch.Tensor:
    # dict-encoded vs scalar-dict single-value fast path
    if isinstance(a, StringColumn) and isinstance(b, StringColumn):
        a_short = (len(a) == 0 or int((a.offsets[1:] - a.offsets[:-1]).max().item()) <= 7) \
            if not a.is_dict else False
        if op in ("=", "!=") and not a.is_dict \
                and b.is_dict and b.dict_size == 1:
            target = b.dict_values()[0]
            if target == "":
                # `col <> ''` / `col = ''`: a length test, not a hash —
                # hashing 100M raw URLs for this cost 150 ms (ClickBench
                # q36/q37)
                m = (a.offsets[1:] - a.offsets[:-1]) == 0
                return m if op == "=" else ~m
            if a.is_cuda or a_short:
                from .joins import raw_string_key

                keys = raw_string_key(a)
                litk = _literal_string_keys(a, [target])
                m = keys == litk[0]
                return m if op == "=" else ~m
        if op in ("=", "!=") and not a.is_dict and not b.is_dict and (a.is_cuda or a_short):
            from .joins import raw_string_key

            # column-vs-column equality via keys (exact <=7B, hashed beyond)
            ka, kb = raw_string_key(a), raw_string_key(b)
            if len(a) == len(b):
                m = ka == kb
                return m if op == "=" else ~m
        if a.is_dict and b.is_dict and len(b) > 0:
            if b.dict_size == 1:
                # b is a broadcast literal
                target = b.dict_values()[0]
                if op in ("=", "!="):
                    code = a.dict_code_of(target)  # O(log n) sorted-dict search
                    if code < 0:
                        m = torch.zeros(len(a), dtype=torch.bool, device=a.device)
                    else:
                        m = a.codes == code
                    return m if op == "=" else ~m
                avals = a.dict_values()
                if op in ("=", "!="):
                    try:
                        code = avals.index(target)
                        m = a.codes == code
                    except ValueError:
                        m = torch.zeros(len(a), dtype=torch.bool, device=a.device)
                    return m if op == "=" else ~m
                # ordering against literal: map dict order
                import numpy as np

                keys = np.array(avals)
                cmp = {"<": keys < target, "<=": keys <= target,
                       ">": keys > target, ">=": keys >= target}[op]
                lut = torch.from_numpy(cmp).to(a.device)
                return lut[a.codes.long()]
        # generic: host comparison (CPU reference path)
        av = a.to_pylist()
        bv = b.to_pylist()
        f = {"=": lambda x, y: x == y, "!=": lambda x, y: x != y,
             "<": lambda x, y: x < y, "<=": lambda x, y: x <= y,
             ">": lambda x, y: x > y, ">=": lambda x, y: x >= y}[op]
        out = [bool(f(x, y)) if x is not None and y is not None else False for x, y in zip(av, bv)]
        return torch.tensor(out, dtype=torch.bool, device=a.device)
    raise EvalError("string comparison with non-string")


def _string_isin(c: StringColumn, lits: List[str]) -> torch.Tensor:
    if c.is_dict:
        vals = c.dict_values()
        hit = torch.tensor([v in lits for v in vals], dtype=torch.bool, device=c.device)
        return hit[c.codes.long().clamp_min(0)] & (c.codes >= 0)
    from .joins import raw_string_key

    lens = c.offsets[1:] - c.offsets[:-1]
    max_len = int(lens.max().item()) if len(c) else 0
    if c.is_cuda or max_len <= 7:
        keys = raw_string_key(c)
        litk = _literal_string_keys(c, lits)
        return torch.isin(keys, litk)
    vals = c.to_pylist()
    return torch.tensor([v in lits if v is not None else False for v in vals],
                        dtype=torch.bool, device=c.device)


_LIKE_CACHE = {}


def like_to_regex(pattern: str) -> "re.Pattern":
    if pattern not in _LIKE_CACHE:
        out = []
        i = 0
        while i < len(pattern):
            ch = pattern[i]
            if ch == "\\" and i + 1 < len(pattern):
                out.append(re.escape(pattern[i + 1]))
                i += 2
                continue
            if ch == "%":
                out.append(".*")
            elif ch == "_":
                out.append(".")
            else:
                out.append(re.escape(ch))
            i += 1
        _LIKE_CACHE[pattern] = re.compile("^" + "".join(out) + "$", re.DOTALL)
    return _LIKE_CACHE[pattern]


def string_like(c: StringColumn, pattern: str, case_insensitive=False, is_regex=False) -> torch.Tensor:
    """LIKE evaluation. Dict columns: evaluate once per dictionary entry and
    gather through codes — O(|dict|) regex work instead of O(n). Raw columns:
    device kernel on GPU (ops/strings), host fallback on CPU."""
    if is_regex:
        rx = re.compile(pattern, re.IGNORECASE if case_insensitive else 0)
        match = lambda s: rx.search(s) is not None
    else:
        rx = like_to_regex(pattern if not case_insensitive else pattern.lower())
        match = (lambda s: rx.match(s.lower()) is not None) if case_insensitive else (
            lambda s: rx.match(s) is not None)
    if c.is_dict:
        # evaluate once per dictionary entry; big dictionaries go through the
        # device kernel (host regex over a 1M-entry dict costs ~0.3 s)
        if c.is_cuda and not is_regex and not case_insensitive:
            from ..ops import kernels as K

            hit = K.require().like_mask(c.offsets, c.bytes_, pattern.encode())
        else:
            vals = c.dict_values()
            hit = torch.tensor([match(v) for v in vals], dtype=torch.bool, device=c.device)
        return hit[c.codes.long().clamp_min(0)] & (c.codes >= 0)
    if c.is_cuda:
        from ..ops import kernels as K

        m = K.like_mask(c, pattern, case_insensitive=case_insensitive, is_regex=is_regex)
        if m is not None:
            return m
    vals = c.to_pylist()
    return torch.tensor([match(v) if v is not None else False for v in vals],
                        dtype=torch.bool, device=c.device)


def _arith(op: str, a: Column, b: Column, out_type: T.DataType) -> Column:
    validity = _merge_validity(a, b)
    at, bt = a.dtype, b.dtype
    if isinstance(out_type, T.DecimalType):
        return _decimal_arith(op, a, b, out_type, validity)
    x, y = a.data, b.data
    if op == "/":
        x = x.to(torch.float64)
        y = y.to(torch.float64)
        data = x / y
        # division by zero -> null (Spark)
        zero = y == 0
        if bool(zero.any()):
            v = validity if validity is not None else torch.ones(len(a), dtype=torch.uint8, device=a.device)
            validity = (v.to(torch.bool) & ~zero).to(torch.uint8)
        return Column(T.F64, data, validity)
    if x.dtype != y.dtype:
        ct = torch.promote_types(x.dtype, y.dtype)
        x, y = x.to(ct), y.to(ct)
    if op == "+":
        data = x + y
    elif op == "-":
        data = x - y
    elif op == "*":
        data = x * y
    elif op == "%":
        # Spark % truncates toward zero (sign of dividend), unlike torch.remainder
        ysafe = torch.where(y != 0, y, torch.ones_like(y))
        if x.dtype.is_floating_point:
            data = x - torch.trunc(x / ysafe) * ysafe
        else:
            data = x - torch.div(x, ysafe, rounding_mode="trunc") * ysafe
        zero = y == 0
        if bool(zero.any()):
            v = validity if validity is not None else torch.ones(len(a), dtype=torch.uint8, device=a.device)
            validity = (v.to(torch.bool) & ~zero).to(torch.uint8)
    elif op == "div":
        data = torch.div(x, y, rounding_mode="trunc").to(torch.int64)
    else:
        raise EvalError(op)
    storage = out_type.storage
    if storage is not None and data.dtype != storage:
        data = data.to(storage)
    return Column(out_type, data, validity)


def _decimal_arith(op: str, a: Column, b: Column, out_type: T.DecimalType, validity) -> Column:
    sa = a.dtype.scale if isinstance(a.dtype, T.DecimalType) else 0
    sb = b.dtype.scale if isinstance(b.dtype, T.DecimalType) else 0
    so = out_type.scale
    x, y = a.data.to(torch.int64), b.data.to(torch.int64)
    if op in ("+", "-"):
        if sa < so:
            x = x * (10 ** (so - sa))
        if sb < so:
            y = y * (10 ** (so - sb))
        data = x + y if op == "+" else x - y
    elif op == "*":
        # result scale = sa + sb; rescale to out scale
        data = x * y
        cur = sa + sb
        data = _rescale_int(data, cur, so)
    elif op == "/":
        xf = x.to(torch.float64) / (10.0 ** sa)
        yf = y.to(torch.float64) / (10.0 ** sb)
        q = xf / yf
        data = torch.round(q * (10.0 ** so)).to(torch.int64)
        zero = y == 0
        if bool(zero.any()):
            base = validity.to(torch.bool) if validity is not None else torch.ones(len(a), dtype=torch.bool, device=a.device)
            validity = (base & ~zero).to(torch.uint8)
    elif op == "%":
        cs = max(sa, sb)
        xs = x * (10 ** (cs - sa))
        ys = y * (10 ** (cs - sb))
        data = torch.where(ys != 0, xs - torch.div(xs, ys, rounding_mode="trunc") * ys, torch.zeros_like(xs))
        data = _rescale_int(data, cs, so)
    else:
        raise EvalError(op)
    return Column(out_type, data, validity)


def _rescale_int(data: torch.Tensor, cur_scale: int, target_scale: int) -> torch.Tensor:
    if cur_scale == target_scale:
        return data
    if cur_scale < target_scale:
        return data * (10 ** (target_scale - cur_scale))
    f = 10 ** (cur_scale - target_scale)
    # round half away from zero (Spark HALF_UP)
    half = f // 2
    adj = torch.where(data >= 0, data + half, data - half)
    return torch.div(adj, f, rounding_mode="trunc")


def _scalar_binop(op: str, l: Scalar, r: Scalar, out_type) -> Scalar:
    if op == "<=>":  # null-safe: never returns null
        if l.is_null or r.is_null:
            return Scalar(l.is_null and r.is_null, T.BOOL)
        return Scalar(l.value == r.value, T.BOOL)
    if l.is_null or r.is_null:
        return Scalar(None, out_type or T.NULL)
    a, b = l.value, r.value
    try:
        res = {
            "+": lambda: a + b, "-": lambda: a - b, "*": lambda: a * b,
            "/": lambda: a / b if b != 0 else None,
            "%": lambda: math.fmod(a, b) if b != 0 else None,
            "div": lambda: int(a / b) if b != 0 else None,
            "=": lambda: a == b, "!=": lambda: a != b, "<": lambda: a < b,
            "<=": lambda: a <= b, ">": lambda: a > b, ">=": lambda: a >= b,
            "<=>": lambda: a == b,
        }[op]()
    except KeyError:
        raise EvalError(f"scalar op {op}")
    return Scalar(res, out_type or T.NULL)


# ---------------------------------------------------------------------------
# Casts
# ---------------------------------------------------------------------------

def cast_value(v: Val, to: T.DataType, chunk: Chunk, try_: bool = False) -> Val:
    if isinstance(v, Scalar):
        if try_:
            try:
                return _cast_scalar(v, to)
            except (ValueError, TypeError, OverflowError):
                return Scalar(None, to)
        return _cast_scalar(v, to)
    return cast_column(v, to)


def _cast_scalar(v: Scalar, to: T.DataType) -> Scalar:
    if v.is_null:
        return Scalar(None, to)
    x = v.value
    if isinstance(to, T.DecimalType):
        return Scalar(float(x), to)  # scaled on broadcast
    if isinstance(to, (T.Int8Type, T.Int16Type, T.Int32Type, T.Int64Type)):
        if isinstance(v.dtype, T.DecimalType):
            return Scalar(int(x), to)
        return Scalar(int(float(x)), to)
    if isinstance(to, (T.Float32Type, T.Float64Type)):
        return Scalar(float(x), to)
    if isinstance(to, T.StringType):
        return Scalar(_value_to_string(x, v.dtype), to)
    if isinstance(to, T.BooleanType):
        if isinstance(x, str):
            return Scalar(x.strip().lower() in ("true", "t", "1", "yes", "y"), to)
        return Scalar(bool(x), to)
    if isinstance(to, T.DateType):
        if isinstance(x, str):
            return Scalar(_date_str_to_days(x), to)
        return Scalar(int(x), to)
    if isinstance(to, T.TimestampType):
        if isinstance(x, str):
            from ..sql.parser import _parse_timestamp_us

            return Scalar(_parse_timestamp_us(x), to)
        return Scalar(int(x), to)
    raise EvalError(f"cast scalar to {to!r}")


def _value_to_string(x, dt: T.DataType) -> str:
    if isinstance(dt, T.DateType):
        d = _dt.date(1970, 1, 1) + _dt.timedelta(days=int(x))
        return d.isoformat()
    if isinstance(dt, T.BooleanType):
        return "true" if x else "false"
    return str(x)


def cast_column(c: Column, to: T.DataType) -> Column:
    frm = c.dtype
    if frm == to:
        return c
    if isinstance(c, StringColumn):
        return _cast_from_string(c, to)
    if isinstance(to, T.StringType):
        return _cast_to_string(c)
    if isinstance(frm, T.DecimalType) and isinstance(to, T.DecimalType):
        return Column(to, _rescale_int(c.data, frm.scale, to.scale), c.validity)
    if isinstance(frm, T.DecimalType):
        if to.is_float:
            return Column(to, c.data.to(torch.float64) / (10.0 ** frm.scale), c.validity)
        if to.is_integer:
            return Column(to, _rescale_int(c.data, frm.scale, 0).to(to.storage), c.validity)
        if isinstance(to, T.BooleanType):
            return Column(to, c.data != 0, c.validity)
    if isinstance(to, T.DecimalType):
        if frm.is_integer or isinstance(frm, T.BooleanType):
            return Column(to, c.data.to(torch.int64) * (10 ** to.scale), c.validity)
        if frm.is_float:
            scaled = torch.round(c.data.to(torch.float64) * (10.0 ** to.scale))
            return Column(to, scaled.to(torch.int64), c.validity)
    if isinstance(frm, T.DateType) and isinstance(to, T.TimestampType):
        return Column(to, c.data.to(torch.int64) * 86_400_000_000, c.validity)
    if isinstance(frm, T.TimestampType) and isinstance(to, T.DateType):
        return Column(to, torch.div(c.data, 86_400_000_000, rounding_mode="floor").to(torch.int32), c.validity)
    if to.storage is not None:
        return Column(to, c.data.to(to.storage), c.validity)
    raise EvalError(f"unsupported cast {frm!r} -> {to!r}")


def _cast_to_string(c: Column) -> StringColumn:
    vals = c.to_pylist()
    out = [None if v is None else _value_to_string(
        (v if not isinstance(c.dtype, T.DateType) else (v - _dt.date(1970, 1, 1)).days), c.dtype)
        for v in vals]
    return StringColumn.from_pylist(out, device=c.device)


def _cast_from_string(c: StringColumn, to: T.DataType) -> Column:
    vals = c.to_pylist()

    def conv(v):
        if v is None:
            return None
        try:
            if isinstance(to, (T.Int8Type, T.Int16Type, T.Int32Type, T.Int64Type)):
                return int(float(v.strip()))
            if to.is_float:
                return float(v.strip())
            if isinstance(to, T.DecimalType):
                return float(v.strip())
            if isinstance(to, T.DateType):
                return _date_str_to_days(v)
            if isinstance(to, T.TimestampType):
                from ..sql.parser import _parse_timestamp_us

                return _parse_timestamp_us(v)
            if isinstance(to, T.BooleanType):
                return v.strip().lower() in ("true", "t", "1", "yes", "y")
        except (ValueError, IndexError):
            return None
        raise EvalError(f"cast string -> {to!r}")

    return Column.from_values([conv(v) for v in vals], to, device=c.device)

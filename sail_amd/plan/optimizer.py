"""Logical optimizer.

Rule pipeline over the resolved plan, mirroring the reference's optimizer
stack (ref: crates/sail-logical-optimizer/src/lib.rs:11,
crates/sail-physical-optimizer/src/lib.rs:1 + join_reorder/):

  1. decorrelate_subqueries — EXISTS/IN -> semi/anti joins; correlated scalar
     aggregate subqueries -> group-by + join (the TPC-H patterns).
  2. pushdown_filters      — split conjunctions, push through project/join.
  3. prune_columns         — drop unreferenced columns below each operator.
  4. join_reorder          — greedy/DP ordering of inner-join chains by
     estimated cardinality (ref: sail-physical-optimizer/src/join_reorder/).
  (constant folding happens inline in the evaluator for scalars)

Rules run on the *resolved* bound tree and must keep schemas/BoundRef indices
consistent; each rule rebuilds indices for changed subtrees.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..engine import types as T
from . import spec as S
from .rules.decorrelate import decorrelate
from .rules.pushdown import pushdown_filters
from .rules.prune import prune_columns
from .rules.thin_agg import thin_aggregates
from .rules.magic_set import semi_filter_aggregates
from .rules.join_order import reorder_joins
from .rules.semi_sink import sink_semi_joins


def optimize(plan: S.Plan, enable_join_reorder: bool = True, stats=None) -> S.Plan:
    plan = decorrelate(plan)
    plan = pushdown_filters(plan)
    if enable_join_reorder:
        plan = reorder_joins(plan, stats)
    plan = sink_semi_joins(plan)
    plan = semi_filter_aggregates(plan, stats)
    plan = thin_aggregates(plan, stats)
    plan = prune_columns(plan)
    return plan

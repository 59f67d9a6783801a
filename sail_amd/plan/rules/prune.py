"""Column pruning (stub — implemented in a later pass)."""
from .. import spec as S


def prune_columns(plan: S.Plan) -> S.Plan:
    return plan

"""Filter pushdown (stub — implemented in a later pass)."""
from .. import spec as S


def pushdown_filters(plan: S.Plan) -> S.Plan:
    return plan

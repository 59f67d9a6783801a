"""Join reordering (stub — implemented in a later pass)."""
from .. import spec as S


def reorder_joins(plan: S.Plan) -> S.Plan:
    return plan

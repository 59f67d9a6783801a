"""Subquery decorrelation (stub — implemented in a later pass)."""
from .. import spec as S


def decorrelate(plan: S.Plan) -> S.Plan:
    return plan

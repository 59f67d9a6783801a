"""sail-mi355x: an MI355X-native Spark-compatible compute engine.

Public surface:
    sail_amd.SessionContext  — build sessions, run SQL
    sail_amd.connect(...)    — Spark Connect server entry (connect/)
"""
from .engine.session import Catalog, DataFrame, SessionContext

__version__ = "0.1.0"

__all__ = ["SessionContext", "DataFrame", "Catalog", "__version__"]

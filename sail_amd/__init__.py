"""sail-mi355x: an MI355X-native Spark-compatible compute engine.

Public surface:
    sail_amd.SessionContext  — build sessions, run SQL
    sail_amd.connect(...)    — Spark Connect server entry (connect/)
"""
from .engine.session import Catalog, DataFrame, SessionContext
from .errors import (AnalysisException, ExecutionException, ParseException,
                     SailError, UnsupportedOperationException)

__version__ = "0.1.0"


def connect_server(host="127.0.0.1", port=0, device=None):
    """Start a Spark Connect server; returns the server (``.address`` holds
    the bound endpoint)."""
    from .connect.server import SparkConnectServer

    return SparkConnectServer(host=host, port=port, device=device).start()


__all__ = ["SessionContext", "DataFrame", "Catalog", "connect_server",
           "SailError", "ParseException", "AnalysisException",
           "ExecutionException", "UnsupportedOperationException",
           "__version__"]

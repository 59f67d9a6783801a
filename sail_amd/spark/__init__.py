"""pysail-compatible surface: `from sail_amd.spark import SparkConnectServer`
(ref: python/pysail/spark/__init__.py:10)."""
from ..connect.server import SparkConnectServer

__all__ = ["SparkConnectServer"]

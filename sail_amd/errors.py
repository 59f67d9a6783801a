"""Structured error hierarchy (Spark-compatible names).

The analogue of the reference's error classes (ref: crates/sail-common
error types surfaced as Spark AnalysisException / ParseException over
Connect). Engine exceptions subclass these so callers can catch by Spark
exception name; each carries a SQLSTATE and optional message parameters.
"""
from __future__ import annotations

from typing import Dict, Optional


class SailError(Exception):
    """Base class for all engine errors."""

    sql_state: str = "XX000"
    error_class: str = "INTERNAL_ERROR"

    def __init__(self, message: str = "",
                 error_class: Optional[str] = None,
                 message_parameters: Optional[Dict[str, str]] = None,
                 sql_state: Optional[str] = None):
        super().__init__(message)
        if error_class is not None:
            self.error_class = error_class
        if sql_state is not None:
            self.sql_state = sql_state
        self.message_parameters = message_parameters or {}

    @property
    def message(self) -> str:
        return str(self)


class ParseException(SailError):
    """SQL could not be parsed (ref: Spark ParseException)."""

    sql_state = "42601"
    error_class = "PARSE_SYNTAX_ERROR"


class AnalysisException(SailError):
    """Plan resolution failed: unknown table/column/function, type errors
    (ref: Spark AnalysisException)."""

    sql_state = "42000"
    error_class = "ANALYSIS_ERROR"


class ExecutionException(SailError):
    """Runtime failure while executing a resolved plan."""

    sql_state = "39000"
    error_class = "EXECUTION_ERROR"


class UnsupportedOperationException(SailError):
    sql_state = "0A000"
    error_class = "UNSUPPORTED_OPERATION"


class TableNotFoundException(AnalysisException):
    error_class = "TABLE_OR_VIEW_NOT_FOUND"
    sql_state = "42P01"


class ColumnNotFoundException(AnalysisException):
    error_class = "UNRESOLVED_COLUMN"
    sql_state = "42703"

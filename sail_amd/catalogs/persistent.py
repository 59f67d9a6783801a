"""File-backed persistent catalog provider.

The reference's catalog layer is an async provider API (create/get/list/
drop for databases/tables/views) with pluggable backends — Glue, HMS,
Iceberg REST, Unity (ref: crates/sail-catalog/src/provider/cache.rs:210-444
and the provider crates). Those need network services the image lacks; this
provider persists the same surface to JSON files so table definitions
survive sessions — the durable-catalog building block the externals plug
into.

Layout: <root>/databases/<db>.json, <root>/tables/<db>.<table>.json with
{"format", "location", "schema", "options", "comment"}.
"""
from __future__ import annotations

import json
import os
import threading
from typing import Dict, List, Optional, Tuple

from ..engine import types as T


class TableDef:
    def __init__(self, name: str, fmt: str, location: str,
                 schema: Optional[List[Tuple[str, T.DataType]]] = None,
                 options: Optional[Dict[str, str]] = None,
                 database: str = "default", comment: str = ""):
        self.name = name
        self.format = fmt
        self.location = location
        self.schema = schema
        self.options = options or {}
        self.database = database
        self.comment = comment

    def to_json(self) -> dict:
        return {
            "name": self.name, "format": self.format,
            "location": self.location,
            "schema": [[n, T.type_name(t)] for n, t in (self.schema or [])],
            "options": self.options, "database": self.database,
            "comment": self.comment,
        }

    @staticmethod
    def from_json(obj: dict) -> "TableDef":
        schema = [(n, T.type_from_name(tn)) for n, tn in obj.get("schema", [])]
        return TableDef(obj["name"], obj["format"], obj["location"],
                        schema or None, obj.get("options"),
                        obj.get("database", "default"),
                        obj.get("comment", ""))


class FileCatalogProvider:
    """The durable provider: same verb surface as the reference's async
    CatalogProvider, synchronous here (local file IO)."""

    def __init__(self, root: str):
        self.root = root
        self._lock = threading.Lock()
        os.makedirs(os.path.join(root, "databases"), exist_ok=True)
        os.makedirs(os.path.join(root, "tables"), exist_ok=True)
        self.create_database("default", if_not_exists=True)

    # -- databases ---------------------------------------------------------
    def _db_path(self, name: str) -> str:
        return os.path.join(self.root, "databases", f"{name.lower()}.json")

    def create_database(self, name: str, if_not_exists: bool = False,
                        comment: str = ""):
        p = self._db_path(name)
        with self._lock:
            if os.path.exists(p):
                if if_not_exists:
                    return
                raise ValueError(f"database {name} already exists")
            with open(p, "w") as f:
                json.dump({"name": name.lower(), "comment": comment}, f)

    def list_databases(self) -> List[str]:
        d = os.path.join(self.root, "databases")
        return sorted(f[:-5] for f in os.listdir(d) if f.endswith(".json"))

    def drop_database(self, name: str, cascade: bool = False):
        with self._lock:
            tables = [t for t in self.list_tables(name)]
            if tables and not cascade:
                raise ValueError(f"database {name} is not empty")
            for t in tables:
                os.remove(self._table_path(name, t))
            p = self._db_path(name)
            if os.path.exists(p):
                os.remove(p)

    # -- tables ------------------------------------------------------------
    def _table_path(self, db: str, name: str) -> str:
        return os.path.join(self.root, "tables",
                            f"{db.lower()}.{name.lower()}.json")

    def create_table(self, td: TableDef, replace: bool = False):
        p = self._table_path(td.database, td.name)
        with self._lock:
            if os.path.exists(p) and not replace:
                raise ValueError(f"table {td.name} already exists")
            tmp = p + ".tmp"
            with open(tmp, "w") as f:
                json.dump(td.to_json(), f)
            os.replace(tmp, p)

    def get_table(self, name: str, db: str = "default") -> Optional[TableDef]:
        p = self._table_path(db, name)
        if not os.path.exists(p):
            return None
        with open(p) as f:
            return TableDef.from_json(json.load(f))

    def list_tables(self, db: str = "default") -> List[str]:
        d = os.path.join(self.root, "tables")
        pre = f"{db.lower()}."
        return sorted(f[len(pre):-5] for f in os.listdir(d)
                      if f.startswith(pre) and f.endswith(".json"))

    def drop_table(self, name: str, db: str = "default",
                   if_exists: bool = False):
        p = self._table_path(db, name)
        with self._lock:
            if not os.path.exists(p):
                if if_exists:
                    return
                raise ValueError(f"table {name} not found")
            os.remove(p)


def attach(session, root: str):
    """Attach a persistent catalog to a session: load every persisted table
    as a scan view and hook create/drop so definitions are durable."""
    from ..plan import spec as S

    provider = FileCatalogProvider(root)
    cat = session.catalog
    cat.persistent = provider
    for db in provider.list_databases():
        for tname in provider.list_tables(db):
            td = provider.get_table(tname, db)
            if td is None:
                continue
            node = S.DataSourceRead(format=td.format, paths=[td.location],
                                    options=dict(td.options))
            if td.schema:
                node.schema = td.schema
            node.__dict__["_table_name"] = tname
            full = tname if db == "default" else f"{db}.{tname}"
            cat.create_view(full, node, replace=True)
    return provider

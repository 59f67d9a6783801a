"""Iceberg REST catalog provider.

Speaks the Iceberg REST Catalog API (the open spec the reference's
`sail-catalog-iceberg` crate generates its client from — ref:
crates/sail-catalog-iceberg/, sail-build-scripts OpenAPI codegen):
`GET /v1/config`, namespace CRUD under `/v1/{prefix}/namespaces`, table
CRUD under `/v1/{prefix}/namespaces/{ns}/tables`. Bearer-token auth.
Same verb surface as `FileCatalogProvider`, so `attach_rest` registers
every table as a scan view exactly like the file-backed catalog.

The image has no network, so the test suite runs this against an
in-process `http.server` fake implementing the same endpoints
(tests/test_storage_catalog.py) — against a real Polaris/Lakekeeper/
Unity endpoint only the base URI changes.
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional
from urllib import error as _uerror
from urllib import request as _urequest

from ..engine import types as T
from .persistent import TableDef

#: multipart namespace separator in URL paths (REST spec: 0x1F)
_NS_SEP = "%1F"


def _iceberg_type(t: T.DataType) -> str:
    if isinstance(t, T.DecimalType):
        return f"decimal({t.precision}, {t.scale})"
    m = {T.BooleanType: "boolean", T.Int32Type: "int", T.Int64Type: "long",
         T.Float32Type: "float", T.Float64Type: "double",
         T.DateType: "date", T.TimestampType: "timestamp",
         T.TimeType: "time", T.BinaryType: "binary",
         T.StringType: "string"}
    for cls, name in m.items():
        if type(t) is cls:
            return name
    if isinstance(t, T.BinaryType):
        return "binary"
    if isinstance(t, T.StringType):
        return "string"
    return "string"


def _engine_type(s: str) -> T.DataType:
    s = s.strip().lower()
    if s.startswith("decimal"):
        inner = s[s.find("(") + 1:s.find(")")]
        p, _, sc = inner.partition(",")
        return T.DecimalType(int(p), int(sc or 0))
    m = {"boolean": T.BOOL, "int": T.I32, "long": T.I64, "float": T.F32,
         "double": T.F64, "date": T.DATE, "timestamp": T.TIMESTAMP,
         "timestamptz": T.TIMESTAMP, "time": T.TIME, "string": T.STRING,
         "binary": T.BINARY, "uuid": T.STRING}
    if s in m:
        return m[s]
    raise ValueError(f"unsupported iceberg type {s!r}")


class IcebergRestCatalogProvider:
    """Synchronous client over the async surface the reference exposes
    (create/get/list/drop for databases=namespaces and tables)."""

    def __init__(self, uri: str, token: Optional[str] = None,
                 warehouse: Optional[str] = None, timeout: float = 10.0):
        self.base = uri.rstrip("/")
        self.token = token
        self.timeout = timeout
        cfg = self._request("GET", "/v1/config" +
                            (f"?warehouse={warehouse}" if warehouse else ""))
        props = dict(cfg.get("defaults", {}))
        props.update(cfg.get("overrides", {}))
        self.properties = props
        prefix = props.get("prefix", "")
        self.prefix = f"/{prefix}" if prefix else ""

    # -- transport ---------------------------------------------------------
    def _request(self, method: str, path: str, body: Optional[dict] = None):
        url = self.base + path
        data = json.dumps(body).encode() if body is not None else None
        req = _urequest.Request(url, data=data, method=method)
        req.add_header("Content-Type", "application/json")
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        try:
            with _urequest.urlopen(req, timeout=self.timeout) as resp:
                raw = resp.read()
                return json.loads(raw) if raw else {}
        except _uerror.HTTPError as e:
            detail = e.read().decode(errors="replace")[:500]
            raise ValueError(
                f"iceberg-rest {method} {path}: HTTP {e.code} {detail}"
            ) from None

    def _ns_path(self, db: str) -> str:
        ns = _NS_SEP.join(db.split("."))
        return f"{self.prefix}/v1/namespaces/{ns}"

    # -- databases (namespaces) -------------------------------------------
    def create_database(self, name: str, if_not_exists: bool = False,
                        comment: str = ""):
        body = {"namespace": name.split("."),
                "properties": ({"comment": comment} if comment else {})}
        try:
            self._request("POST", f"{self.prefix}/v1/namespaces", body)
        except ValueError as e:
            if if_not_exists and "409" in str(e):
                return
            raise

    def list_databases(self) -> List[str]:
        out = self._request("GET", f"{self.prefix}/v1/namespaces")
        return sorted(".".join(ns) for ns in out.get("namespaces", []))

    def drop_database(self, name: str, cascade: bool = False):
        if cascade:
            for t in self.list_tables(name):
                self.drop_table(t, name, if_exists=True)
        self._request("DELETE", self._ns_path(name))

    # -- tables ------------------------------------------------------------
    def create_table(self, td: TableDef, replace: bool = False):
        fields = [{"id": i + 1, "name": n, "type": _iceberg_type(t),
                   "required": False}
                  for i, (n, t) in enumerate(td.schema or [])]
        body = {"name": td.name,
                "schema": {"type": "struct", "schema-id": 0,
                           "fields": fields},
                "properties": dict(td.options)}
        if td.location:
            body["location"] = td.location
        if replace:
            try:
                self.drop_table(td.name, td.database, if_exists=True)
            except ValueError:
                pass
        self._request("POST", self._ns_path(td.database) + "/tables", body)

    def get_table(self, name: str, db: str = "default"
                  ) -> Optional[TableDef]:
        try:
            out = self._request(
                "GET", self._ns_path(db) + f"/tables/{name}")
        except ValueError as e:
            if "404" in str(e):
                return None
            raise
        md = out.get("metadata", {})
        schemas = md.get("schemas") or ([md["schema"]] if "schema" in md
                                        else [])
        cur = md.get("current-schema-id", 0)
        schema = None
        for sc in schemas:
            if sc.get("schema-id", 0) == cur or len(schemas) == 1:
                schema = [(f["name"], _engine_type(f["type"]))
                          for f in sc.get("fields", [])
                          if isinstance(f.get("type"), str)]
        return TableDef(name, "iceberg", md.get("location", ""),
                        schema or None, md.get("properties", {}), db)

    def list_tables(self, db: str = "default") -> List[str]:
        out = self._request("GET", self._ns_path(db) + "/tables")
        return sorted(ident["name"] for ident in out.get("identifiers", []))

    def drop_table(self, name: str, db: str = "default",
                   if_exists: bool = False):
        try:
            self._request("DELETE", self._ns_path(db) + f"/tables/{name}")
        except ValueError as e:
            if if_exists and "404" in str(e):
                return
            raise


def attach_rest(session, uri: str, token: Optional[str] = None,
                warehouse: Optional[str] = None):
    """Register every REST-catalog table as a scan view on the session
    (mirror of persistent.attach for the network-backed provider)."""
    from ..plan import spec as S

    provider = IcebergRestCatalogProvider(uri, token=token,
                                          warehouse=warehouse)
    cat = session.catalog
    cat.rest = provider
    for db in provider.list_databases():
        for tname in provider.list_tables(db):
            td = provider.get_table(tname, db)
            if td is None or not td.location:
                continue
            node = S.DataSourceRead(format=td.format, paths=[td.location],
                                    options=dict(td.options))
            if td.schema:
                node.schema = td.schema
            node.__dict__["_table_name"] = tname
            full = tname if db == "default" else f"{db}.{tname}"
            cat.create_view(full, node, replace=True)
    return provider

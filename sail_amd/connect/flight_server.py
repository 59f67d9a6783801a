"""Arrow Flight server: SQL in, Arrow record batches out.

The role of the reference's sail-flight service
(ref: crates/sail-flight/src/service.rs:33 SailFlightSqlService —
do_get_statement executes through the same planning path as Spark Connect).
Tickets and flight descriptors carry the SQL statement text; results stream
as Arrow record batches over gRPC.
"""
from __future__ import annotations

import threading
from typing import Optional

import pyarrow as pa
import pyarrow.flight as flight

from ..engine.session import SessionContext


class SailFlightServer(flight.FlightServerBase):
    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 device: Optional[str] = None):
        location = f"grpc://{host}:{port}"
        super().__init__(location)
        self._session = SessionContext(device=device)
        self._host = host

    @property
    def address(self) -> str:
        return f"grpc://{self._host}:{self.port}"

    @property
    def session(self) -> SessionContext:
        return self._session

    # -- Flight RPCs -------------------------------------------------------
    def get_flight_info(self, context, descriptor):
        sql = descriptor.command.decode("utf-8")
        df = self._session.sql(sql)
        table = df.to_arrow()
        endpoint = flight.FlightEndpoint(sql.encode("utf-8"),
                                         [flight.Location(self.address)])
        return flight.FlightInfo(table.schema, descriptor, [endpoint],
                                 table.num_rows, -1)

    def do_get(self, context, ticket):
        sql = ticket.ticket.decode("utf-8")
        df = self._session.sql(sql)
        table = df.to_arrow()
        return flight.RecordBatchStream(table)

    def do_put(self, context, descriptor, reader, writer):
        """Upload a table: descriptor command is the table name."""
        name = descriptor.command.decode("utf-8")
        table = reader.read_all()
        from ..datasource.arrow_io import arrow_to_table

        self._session.catalog.register_table(name, arrow_to_table(table))

    def do_action(self, context, action):
        if action.type == "sql":
            self._session.sql(action.body.to_pybytes().decode("utf-8")).collect_chunk()
            return []
        raise NotImplementedError(action.type)

    def list_flights(self, context, criteria):
        for name in self._session.catalog.list_tables():
            desc = flight.FlightDescriptor.for_command(f"SELECT * FROM {name}".encode())
            sch = self._session.catalog.table_schema(name)
            if sch is None:
                continue
            from ..datasource.arrow_io import _arrow_type

            schema = pa.schema([(n, _arrow_type(t)) for n, t in sch])
            yield flight.FlightInfo(schema, desc, [], -1, -1)


def start_flight_server(host="127.0.0.1", port=0, device=None) -> SailFlightServer:
    srv = SailFlightServer(host=host, port=port, device=device)
    t = threading.Thread(target=srv.serve, daemon=True)
    t.start()
    return srv

"""Arrow Flight server round-trips (ref: crates/sail-flight)."""
import pyarrow as pa
import pyarrow.flight as flight
import pytest

from sail_amd.connect.flight_server import start_flight_server


@pytest.fixture(scope="module")
def server():
    srv = start_flight_server(device="cpu")
    yield srv
    srv.shutdown()


def test_flight_sql_do_get(server):
    server.session.create_dataframe({"a": [1, 2, 3]}, name="t")
    client = flight.connect(server.address)
    reader = client.do_get(flight.Ticket(b"SELECT a * 2 AS x FROM t ORDER BY a"))
    table = reader.read_all()
    assert table.to_pydict() == {"x": [2, 4, 6]}


def test_flight_info_and_put(server):
    client = flight.connect(server.address)
    up = pa.table({"k": ["a", "b"], "v": [1.5, 2.5]})
    desc = flight.FlightDescriptor.for_command(b"uploaded")
    writer, _ = client.do_put(desc, up.schema)
    writer.write_table(up)
    writer.close()
    info = client.get_flight_info(
        flight.FlightDescriptor.for_command(b"SELECT sum(v) AS s FROM uploaded"))
    assert info.total_records == 1
    out = client.do_get(info.endpoints[0].ticket).read_all()
    assert out.to_pydict()["s"] == [4.0]

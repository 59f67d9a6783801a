"""Start the Spark Connect server and drive it with the in-repo client —
SQL and DataFrame-API (relation-tree) round trips over real gRPC.

    python examples/connect_client_demo.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sail_amd
from sail_amd.connect.client import ConnectClient, E, R


def main():
    server = sail_amd.connect_server(port=0)
    client = ConnectClient(server.address)
    print("spark version:", client.spark_version())
    print("sql:", client.sql("SELECT 1 AS one, 'x' AS s").to_pydict())
    sess = server.session(client.session_id)
    sess.create_dataframe({"k": ["a", "b", "a"], "v": [1, 2, 3]}, name="t")
    rel = R.aggregate(R.read_table("t"), group=[E.col("k")],
                      aggs=[E.alias(E.fn("sum", E.col("v")), "sv")])
    print("relation:", client.execute_relation(rel).to_pydict())
    server.stop()


if __name__ == "__main__":
    main()

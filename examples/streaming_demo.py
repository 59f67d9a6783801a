"""Structured-streaming micro-batch demo: file source -> windowed counts
-> parquet sink with offset-WAL recovery.

    python examples/streaming_demo.py
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sail_amd


def main():
    root = tempfile.mkdtemp()
    src = os.path.join(root, "in")
    os.makedirs(src)
    s = sail_amd.SessionContext(device="cpu")
    import json

    with open(os.path.join(src, "f0.json"), "w") as f:
        for row in ({"k": "a", "v": 1}, {"k": "b", "v": 2}):
            f.write(json.dumps(row) + "\n")
    q = (s.read_stream.format("json").load(src, name="events")
         .sql("SELECT k, sum(v) AS sv FROM events GROUP BY k")
         .write_stream.output_mode("complete").format("memory")
         .query_name("demo")
         .option("checkpointLocation", os.path.join(root, "ckpt"))
         .start())
    q.process_all_available()
    print(s.sql("SELECT * FROM demo ORDER BY k").collect())
    q.stop()


if __name__ == "__main__":
    main()

"""Delta + Iceberg tour: create, upsert, time travel, vacuum, REST catalog.

    python examples/lakehouse_tour.py /tmp/lakehouse_demo
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import sail_amd


def main():
    root = sys.argv[1] if len(sys.argv) > 1 else tempfile.mkdtemp()
    s = sail_amd.SessionContext(device="cpu")
    delta = os.path.join(root, "events_delta")
    ice = os.path.join(root, "events_iceberg")

    s.create_dataframe({"id": [1, 2, 3], "v": ["a", "b", "c"]}, name="src")
    s.sql(f"CREATE TABLE delta.`{delta}` AS SELECT * FROM src")
    s.sql(f"MERGE INTO delta.`{delta}` t USING "
          "(SELECT 3 AS id, 'c2' AS v UNION ALL SELECT 4, 'd') u "
          "ON t.id = u.id "
          "WHEN MATCHED THEN UPDATE SET v = u.v "
          "WHEN NOT MATCHED THEN INSERT (id, v) VALUES (u.id, u.v)")
    print("delta now:",
          s.sql(f"SELECT * FROM delta.`{delta}` ORDER BY id").collect())
    print("delta v0 :",
          s.sql(f"SELECT * FROM delta.`{delta}` VERSION AS OF 0 "
                "ORDER BY id").collect())

    s.sql(f"CREATE TABLE iceberg.`{ice}` AS SELECT * FROM src")
    s.sql(f"INSERT INTO iceberg.`{ice}` VALUES (9, 'z')")
    print("iceberg  :",
          s.sql(f"SELECT count(*) FROM iceberg.`{ice}`").collect())
    print("iceberg v0:",
          s.sql(f"SELECT count(*) FROM iceberg.`{ice}` VERSION AS OF 0"
                ).collect())


if __name__ == "__main__":
    main()

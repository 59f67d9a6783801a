"""Delta Lake benchmark workload (BASELINE config #5: scan + MERGE INTO).

Setup writes TPC-H orders (at the given SF, this rank's shard) to a Delta
table on local disk, and registers an `updates` source (~1% of keys with
changed values + new keys). One step = a full-scan aggregate over the Delta
table + a MERGE INTO applying the updates (the MERGE commits a new version;
steps keep appending history like a production upsert loop).
"""
from __future__ import annotations

import os
import tempfile

import torch

from ..engine import types as T
from ..engine.column import Column, Table
from .tpch import TpchGenerator, _gen, _randint

_STATE = {}


def setup_delta_bench(session, sf: float = 10.0, device="cpu", rank: int = 0, world: int = 1):
    gen = TpchGenerator(sf=sf, device=device, rank=rank, world=world)
    orders = gen.orders()
    root = os.environ.get("SAIL_DELTA_BENCH_DIR", tempfile.mkdtemp(prefix="sail_delta_"))
    path = os.path.join(root, f"orders_r{rank}")
    session.catalog.register_table("orders", orders, replicated=(world == 1),
                                   global_rows=gen.n_orders)
    session.table("orders").write.format("delta").mode("overwrite").save(path)

    # updates: ~1% existing keys + 0.1% new keys
    n = orders.num_rows
    g = _gen(7, f"delta_updates{rank}", torch.device(device))
    n_upd = max(1, n // 100)
    keys = orders.columns["o_orderkey"].data
    upd_keys = keys[_randint(0, n - 1, n_upd, g, torch.device(device))]
    new_keys = torch.arange(gen.n_orders + rank * n_upd + 1,
                            gen.n_orders + rank * n_upd + max(1, n // 1000) + 1,
                            dtype=torch.int64, device=torch.device(device))
    allk = torch.unique(torch.cat([upd_keys, new_keys]))
    nu = allk.shape[0]
    updates = Table({
        "o_orderkey": Column(T.I64, allk),
        "o_totalprice": Column(T.DecimalType(12, 2), _randint(100000, 50000000, nu, g, torch.device(device))),
    })
    session.catalog.register_table("updates", updates, replicated=(world == 1))
    if world > 1 and getattr(session, "dist", None) is not None:
        from ..exec.distributed import sync_table_stats

        sync_table_stats(session)
    _STATE["path"] = path
    session.conf["sail.delta.bench.path"] = path


DELTA_QUERIES = {
    1: None,  # placeholders; resolved at runtime via _delta_sql
    2: None,
}


def _delta_sql(session, qid: int) -> str:
    path = session.conf["sail.delta.bench.path"]
    if qid == 1:
        return (f"SELECT o_orderstatus, count(*), sum(o_totalprice), avg(o_totalprice) "
                f"FROM delta.`{path}` GROUP BY o_orderstatus ORDER BY o_orderstatus")
    return (f"MERGE INTO delta.`{path}` AS t USING updates AS u "
            f"ON t.o_orderkey = u.o_orderkey "
            f"WHEN MATCHED THEN UPDATE SET o_totalprice = u.o_totalprice "
            f"WHEN NOT MATCHED THEN INSERT (o_orderkey, o_totalprice) "
            f"VALUES (u.o_orderkey, u.o_totalprice)")

"""Gang-restart failure recovery: a rank dying
mid-suite must not lose the workload — the supervisor reforms the gang and
resumes from the WAL of completed queries."""
import json
import os

import pytest

from sail_amd.exec.failover import GangSupervisor, load_completed

# -- workload hooks (must be importable from spawned processes) -------------

_CRASH_FLAG = None  # set via env in the spawned procs


def gang_setup(rank, world, dist):
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.exec.context import DistContext

    s = sail_amd.SessionContext(device="cpu")
    s.dist = DistContext(dist, rank=rank, world=world, device="cpu")
    register_tpch(s, sf=0.01, rank=rank, world=world)
    return s


def gang_work(session, item, rank, world, attempt):
    from sail_amd.datagen.tpch_queries import QUERIES

    crash_at = os.environ.get("SAIL_TEST_CRASH_ITEM")
    crash_rank = int(os.environ.get("SAIL_TEST_CRASH_RANK", "1"))
    always = os.environ.get("SAIL_TEST_CRASH_ALWAYS") == "1"
    if crash_at == item and rank == crash_rank and (attempt == 0 or always):
        os._exit(17)  # simulate a hard rank death mid-query
    rows = session.sql(QUERIES[int(item)]).collect()
    return {"rows": len(rows), "first": repr(rows[0]) if rows else None}


def test_gang_restart_completes_after_rank_death(tmp_path, monkeypatch):
    monkeypatch.setenv("SAIL_TEST_CRASH_ITEM", "6")
    monkeypatch.setenv("SAIL_TEST_CRASH_RANK", "1")
    sup = GangSupervisor(
        world=2, run_dir=str(tmp_path / "gang"),
        setup="test_failover:gang_setup", work="test_failover:gang_work",
        max_attempts=3)
    items = ["1", "3", "6", "13"]
    done = sup.run(items)
    # the gang died at q6 on attempt 0 and was restarted once
    assert sup.attempts_used == 2
    assert sorted(done) == sorted(items)
    # q1/q3 completed before the crash and were NOT re-run from scratch:
    # their WAL entries survive and every item has a result payload
    for item in items:
        with open(done[item]) as f:
            rec = json.load(f)
        assert rec["rows"] > 0
    # results match a plain single-process run
    import sail_amd
    from sail_amd.datagen.tpch import register_tpch
    from sail_amd.datagen.tpch_queries import QUERIES
    from sail_amd.datagen.tpch import TpchGenerator
    from sail_amd.engine.executor import concat_columns
    from sail_amd.engine.column import Table

    single = sail_amd.SessionContext(device="cpu")
    shards = [TpchGenerator(sf=0.01, device="cpu", rank=r, world=2).generate_all()
              for r in range(2)]
    for name in shards[0]:
        if name in ("region", "nation"):
            single.catalog.register_table(name, shards[0][name])
            continue
        cols = {cn: concat_columns([shards[r][name].columns[cn]
                                    for r in range(2)])
                for cn in shards[0][name].columns}
        single.catalog.register_table(name, Table(cols))
    for item in items:
        want = single.sql(QUERIES[int(item)]).collect()
        with open(done[item]) as f:
            rec = json.load(f)
        assert rec["rows"] == len(want)
        if want:
            assert rec["first"] == repr(want[0])


def test_gang_gives_up_after_max_attempts(tmp_path, monkeypatch):
    monkeypatch.setenv("SAIL_TEST_CRASH_ITEM", "1")
    monkeypatch.setenv("SAIL_TEST_CRASH_RANK", "0")
    monkeypatch.setenv("SAIL_TEST_CRASH_ALWAYS", "1")
    sup = GangSupervisor(
        world=2, run_dir=str(tmp_path / "gang2"),
        setup="test_failover:gang_setup", work="test_failover:gang_work",
        max_attempts=2)
    with pytest.raises(RuntimeError, match="after 2 attempts"):
        sup.run(["1"])
    assert sup.attempts_used == 2


def test_wal_resume_skips_completed(tmp_path):
    run = str(tmp_path / "wal")
    os.makedirs(run)
    with open(os.path.join(run, "completed.wal"), "w") as f:
        f.write(json.dumps({"item": "1", "result": ""}) + "\n")
    done = load_completed(run)
    assert done == {"1": ""}

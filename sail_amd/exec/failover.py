"""Gang-restart supervision for SPMD query execution.

The reference retries failed tasks at *region* granularity on its
driver/worker cluster (ref: crates/sail-execution/src/driver/job_scheduler/
core.rs:154-270, cluster.task_max_attempts). An SPMD engine has no task
regions — every rank executes the same plan — so the retry unit here is the
QUERY (or micro-batch): a host-side supervisor launches one process per
rank, rank 0 journals completed work items to a WAL, and when any rank
dies the whole gang is torn down, the communicator re-formed on a fresh
port, and execution resumes from the first unjournaled item, up to
`max_attempts` gang incarnations.

This is the CPU/gloo-testable core; on a GPU node the same supervisor runs
with backend="nccl" (RCCL), one rank per GPU.
"""
from __future__ import annotations

import json
import os
import socket
import time
from typing import Callable, Dict, List, Optional

import torch.multiprocessing as mp


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _wal_path(run_dir: str) -> str:
    return os.path.join(run_dir, "completed.wal")


def load_completed(run_dir: str) -> Dict[str, str]:
    """WAL of completed items: {item_id: result_path}."""
    out: Dict[str, str] = {}
    p = _wal_path(run_dir)
    if not os.path.exists(p):
        return out
    with open(p) as f:
        for line in f:
            if line.strip():
                rec = json.loads(line)
                out[rec["item"]] = rec.get("result", "")
    return out


def _gang_worker(rank: int, world: int, port: int, run_dir: str,
                 items: List[str], setup_name: str, work_name: str,
                 attempt: int):
    """One rank of the gang: init process group, run the remaining items in
    order, journal completions (rank 0) after a barrier so an item is only
    WAL-committed when EVERY rank finished it."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import importlib

    import torch.distributed as dist

    mod_name, _, fn = setup_name.rpartition(":")
    setup_fn = getattr(importlib.import_module(mod_name), fn)
    mod_name, _, fn = work_name.rpartition(":")
    work_fn = getattr(importlib.import_module(mod_name), fn)

    backend = os.environ.get("SAIL_GANG_BACKEND", "gloo")
    dist.init_process_group(backend, rank=rank, world_size=world)
    try:
        ctx = setup_fn(rank, world, dist)
        done = load_completed(run_dir)
        for item in items:
            if item in done:
                continue
            result = work_fn(ctx, item, rank=rank, world=world,
                            attempt=attempt)
            dist.barrier()  # all ranks finished this item
            if rank == 0:
                rec = {"item": item, "ts": time.time()}
                if result is not None:
                    rp = os.path.join(run_dir, f"result-{item}.json")
                    with open(rp, "w") as f:
                        json.dump(result, f)
                    rec["result"] = rp
                with open(_wal_path(run_dir), "a") as f:
                    f.write(json.dumps(rec) + "\n")
                    f.flush()
                    os.fsync(f.fileno())
            dist.barrier()  # WAL visible before anyone starts the next item
    finally:
        try:
            dist.destroy_process_group()
        except Exception:
            pass


class GangSupervisor:
    """Launch + supervise an SPMD gang; gang-restart on any rank failure.

    setup/work are importable-name strings ("pkg.mod:fn") so spawned
    processes can resolve them:
      setup(rank, world, dist) -> ctx
      work(ctx, item, rank=, world=, attempt=) -> json-serializable | None
    """

    def __init__(self, world: int, run_dir: str, setup: str, work: str,
                 max_attempts: int = 3, join_timeout: float = 300.0):
        self.world = world
        self.run_dir = run_dir
        self.setup = setup
        self.work = work
        self.max_attempts = max_attempts
        self.join_timeout = join_timeout
        self.attempts_used = 0
        os.makedirs(run_dir, exist_ok=True)

    def run(self, items: List[str]) -> Dict[str, str]:
        ctx = mp.get_context("spawn")
        last_err: Optional[str] = None
        for attempt in range(self.max_attempts):
            self.attempts_used = attempt + 1
            done = load_completed(self.run_dir)
            if all(i in done for i in items):
                return done
            port = _free_port()
            procs = [ctx.Process(
                target=_gang_worker,
                args=(r, self.world, port, self.run_dir, items,
                      self.setup, self.work, attempt))
                for r in range(self.world)]
            for p in procs:
                p.start()
            failed = False
            deadline = time.time() + self.join_timeout
            alive = list(procs)
            while alive and time.time() < deadline:
                for p in list(alive):
                    p.join(timeout=0.05)
                    if p.exitcode is None:
                        continue
                    alive.remove(p)
                    if p.exitcode != 0:
                        # one rank died: tear the gang down (the reference
                        # cancels the whole region's attempts the same way)
                        failed = True
                        last_err = f"rank exited {p.exitcode}"
                        for q in procs:
                            if q.is_alive():
                                q.terminate()
                        for q in procs:
                            q.join(timeout=30)
                        alive = []
                        break
            if alive:  # timed out
                failed = True
                last_err = "gang join timeout"
                for q in procs:
                    if q.is_alive():
                        q.terminate()
                for q in procs:
                    q.join(timeout=30)
            if not failed:
                done = load_completed(self.run_dir)
                if all(i in done for i in items):
                    return done
                failed = True
                last_err = "gang exited without completing all items"
        raise RuntimeError(
            f"gang failed after {self.max_attempts} attempts: {last_err}")

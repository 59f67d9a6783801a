"""A/B the grouped MIN/MAX strategies at ClickBench q21 shape.

q21 (0-based) spends ~700 ms in a 120k-group, 2-agg pass over ~100M rows;
the suspect is scatter_reduce_(amin) (round-1 profiles showed ROCm
scatter_reduce pathologically slow at high contention). Candidates:
  A. scatter_reduce_(amin)            (current aggregates.py path)
  B. sort packed (gid*S + val), segment-first  (rocPRIM radix underneath)
  C. (REMOVED) sort by val + scatter_ "last write wins": INVALID on GPU —
     scatter_ with duplicate indices has unspecified write order, measured
     wrong on MI355X (the assert caught it). Kept here as the record.
Measured on MI355X (2026-09-12): the aggregate path is no longer a
bottleneck after the dict lex-sort fix (ClickBench q21 736->40 ms), so
the scatter_reduce path stays.
Run: python tools/bench_minmax.py [n_rows] [n_groups] [val_span]
"""
import sys
import time

import torch


def timeit(fn, reps=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / reps * 1000


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 100_000_000
    ng = int(sys.argv[2]) if len(sys.argv) > 2 else 120_000
    span = int(sys.argv[3]) if len(sys.argv) > 3 else 1_200_000
    dev = "cuda:0"
    g = torch.Generator(device=dev).manual_seed(7)
    gid = torch.randint(0, ng, (n,), device=dev, generator=g)
    vals = torch.randint(0, span, (n,), device=dev, generator=g)

    def a_scatter_reduce():
        out = torch.full((ng,), 2**62, dtype=torch.int64, device=dev)
        out.scatter_reduce_(0, gid, vals, reduce="amin", include_self=True)
        return out

    def b_sort_packed():
        packed = gid * span + vals
        s, _ = torch.sort(packed)
        gs = s // span
        first = torch.ones(n, dtype=torch.bool, device=dev)
        first[1:] = gs[1:] != gs[:-1]
        out = torch.full((ng,), 2**62, dtype=torch.int64, device=dev)
        out[gs[first]] = s[first] - gs[first] * span
        return out

    ra, rb = a_scatter_reduce(), b_sort_packed()
    assert torch.equal(ra, rb), "B mismatch"
    print(f"n={n} ng={ng} span={span}")
    print(f"A scatter_reduce amin : {timeit(a_scatter_reduce):8.2f} ms")
    print(f"B sort packed         : {timeit(b_sort_packed):8.2f} ms")

    # the other q21 pieces at the same shape, for the 706 ms budget
    def count():
        out = torch.zeros(ng, dtype=torch.int64, device=dev)
        out.index_add_(0, gid, torch.ones(n, dtype=torch.int64, device=dev))
        return out

    from sail_amd.engine.aggregates import group_ids
    from sail_amd.engine.column import Column
    from sail_amd.engine import types as T

    def gids():
        return group_ids([Column(T.I64, gid)])

    print(f"count index_add       : {timeit(count):8.2f} ms")
    print(f"group_ids dense       : {timeit(lambda: gids(), 3):8.2f} ms")


if __name__ == "__main__":
    main()

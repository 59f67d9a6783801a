"""A/B the grouped MIN/MAX strategies at ClickBench q21 shape.

q21 (0-based) spends ~700 ms in a 120k-group, 2-agg pass over ~100M rows;
the suspect is scatter_reduce_(amin) (round-1 profiles showed ROCm
scatter_reduce pathologically slow at high contention). Candidates:
  A. scatter_reduce_(amin)            (current aggregates.py path)
  B. sort packed (gid*S + val), segment-first  (rocPRIM radix underneath)
  C. index_put-free: sort by val once, scatter_ winners (last write wins
     on a descending-sorted value order => min)
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

    def c_sort_vals_scatter():
        order = torch.argsort(vals, descending=True)
        out = torch.full((ng,), 2**62, dtype=torch.int64, device=dev)
        out.scatter_(0, gid.index_select(0, order),
                     vals.index_select(0, order))
        return out

    ra, rb, rc = a_scatter_reduce(), b_sort_packed(), c_sort_vals_scatter()
    assert torch.equal(ra, rb), "B mismatch"
    assert torch.equal(ra, rc), "C mismatch"
    print(f"n={n} ng={ng} span={span}")
    print(f"A scatter_reduce amin : {timeit(a_scatter_reduce):8.2f} ms")
    print(f"B sort packed         : {timeit(b_sort_packed):8.2f} ms")
    print(f"C sort vals + scatter : {timeit(c_sort_vals_scatter):8.2f} ms")

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

"""grouped_acc throughput probe: python tools/agg_microbench.py [n]"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from sail_amd.ops import kernels as K

n = int(sys.argv[1]) if len(sys.argv) > 1 else 600_000_000
ext = K.require()
dev = "cuda"
gid4 = torch.randint(0, 4, (n,), dtype=torch.int32, device=dev)
gid8 = torch.randint(0, 8, (n,), dtype=torch.int32, device=dev)
vals = [torch.randint(0, 1 << 30, (n,), dtype=torch.int64, device=dev) for _ in range(4)]
mask = (torch.rand(n, device=dev) < 0.98)


def bench(name, fn, bytes_moved, iters=4):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{name:<34} {dt*1000:8.2f} ms  {bytes_moved/dt/1e9:7.0f} GB/s")


by = n * (4 + 1 + 4 * 8)
bench("grouped_acc G=4 NC=4 sum_i64", lambda: ext.grouped_acc(gid4, mask, vals, [0, 0, 0, 0], 4), by)
bench("grouped_acc G=8 NC=4 sum_i64", lambda: ext.grouped_acc(gid8, mask, vals, [0, 0, 0, 0], 8), by)
bench("grouped_acc G=4 NC=4 sum_f64", lambda: ext.grouped_acc(gid4, mask, vals, [1, 1, 1, 1], 4), by)
bench("grouped_acc G=4 NC=2", lambda: ext.grouped_acc(gid4, mask, vals[:2], [0, 0], 4), n * (4 + 1 + 16))
bench("grouped_acc G=4 NC=1", lambda: ext.grouped_acc(gid4, mask, vals[:1], [0], 4), n * 13)
# torch baseline: one index_add per column
out = torch.zeros(4, dtype=torch.int64, device=dev)
bench("torch index_add x4 (G=4)", lambda: [torch.zeros(4, dtype=torch.int64, device=dev).index_add_(0, gid4.long(), v) for v in vals], by + n * 8)

"""Microbenchmark of the primitive device ops the executor leans on.

Prints achieved GB/s for each; run on the GPU box to calibrate which engine
steps are at the HBM roofline and which need custom kernels.
"""
import sys
import time

import torch

sys.path.insert(0, ".")


def bench(name, bytes_moved, fn, iters=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{name:<42} {dt*1000:8.2f} ms   {bytes_moved/dt/1e9:8.0f} GB/s")


def main():
    n = 600_000_000
    dev = "cuda"
    a = torch.randint(0, 1 << 40, (n,), dtype=torch.int64, device=dev)
    b = torch.randint(1, 100, (n,), dtype=torch.int64, device=dev)
    a32 = a.to(torch.int32)
    f = a.to(torch.float64)
    mask = b > 50

    bench("int64 mul (r16W8)", n * 24, lambda: a * b)
    bench("int64 add scalar (r8W8)", n * 16, lambda: a + 7)
    bench("int64 cmp scalar -> bool (r8W1)", n * 9, lambda: a > (1 << 39))
    bench("bool and (r2W1)", n * 3, lambda: mask & mask)
    bench("int64 zeros (W8)", n * 8, lambda: torch.zeros(n, dtype=torch.int64, device=dev))
    bench("int64->int32 cast (r8W4)", n * 12, lambda: a.to(torch.int32))
    bench("abs.max reduce (r8)", n * 8, lambda: a.abs().max())
    bench("masked gather idx (nonzero)", n * 9, lambda: torch.nonzero(mask))
    bench("gather int64 by idx[n/2]", n // 2 * 16 + n // 2 * 8,
          lambda: a.index_select(0, torch.arange(0, n, 2, device=dev)))
    bench("torch.where (r17W8)", n * 25, lambda: torch.where(mask, a, b))

    from sail_amd.ops import kernels as K

    ext = K.require()
    gid = torch.zeros(n, dtype=torch.int32, device=dev)
    bench("grouped_acc 1col sum (r8+4W0)", n * 12,
          lambda: ext.grouped_acc(gid, None, [a], [0], 4))
    bench("grouped_acc 2col+mask (r8+8+4+1)", n * 21,
          lambda: ext.grouped_acc(gid, mask, [a, b], [0, 0], 4))
    # q6-like pipeline
    disc = torch.randint(0, 11, (n,), dtype=torch.int64, device=dev)
    qty = torch.randint(100, 5100, (n,), dtype=torch.int64, device=dev)
    ship = torch.randint(8000, 10600, (n,), dtype=torch.int32, device=dev)

    def q6_like():
        m = (ship >= 8766) & (ship < 9131) & (disc >= 5) & (disc <= 7) & (qty < 2400)
        v = a * disc
        return ext.grouped_acc(torch.zeros(n, dtype=torch.int32, device=dev), m, [v], [0], 1)

    bench("q6-like pipeline", n * 60, q6_like)


if __name__ == "__main__":
    main()

#!/usr/bin/env python3
"""Bandwidth microbenchmark for the gfx950 flat-arena kernels.

Each op's achieved GB/s vs the MI355X HBM3E roof (~8 TB/s) at several
arena sizes.  Bytes counted = actual global traffic (reads + writes).
"""

import json
import sys
import time

import torch

sys.path.insert(0, ".")
from msrflute_amd import ops  # noqa: E402


def timeit(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n


def main():
    assert torch.cuda.is_available() and ops.HAS_EXT
    results = {}
    for n in (1 << 20, 1 << 24, 1 << 28):
        x = torch.randn(n, device="cuda")
        y = torch.randn(n, device="cuda")
        m = torch.zeros(n, device="cuda")
        v = torch.zeros(n, device="cuda")
        stats = torch.zeros(2, device="cuda")
        lr_t = torch.full((1,), 0.01, device="cuda")
        fb = 4 * n / 1e9  # GB per full pass over one buffer

        def gbs(sec, nbuf):
            return nbuf * fb / sec

        r = {}
        r["axpy (2R+1W)"] = gbs(timeit(lambda: ops.axpy(y, x, 0.5)), 3)
        r["scale (1R+1W)"] = gbs(timeit(lambda: ops.scale(x, 1.0001)), 2)
        r["sum_sumsq (1R)"] = gbs(timeit(lambda: ops.sum_sumsq(x)), 1)
        r["clip_stats (2R+1W)"] = gbs(
            timeit(lambda: ops.clip_stats_accumulate(x, 1e9, stats)), 3)
        r["pseudo_grad (2R+1W)"] = gbs(
            timeit(lambda: ops.pseudo_grad(m, x, y, 1.0)), 3)
        r["sgd_step (2R+1W)"] = gbs(
            timeit(lambda: ops.sgd_step_devlr(x, y, None, lr_t)), 3)
        r["adam_step (4R+3W)"] = gbs(
            timeit(lambda: ops.adam_step(x, y, m, v, None, step=2, lr=1e-3)),
            7)
        r["gauss_noise (1R+1W)"] = gbs(
            timeit(lambda: ops.add_gaussian_noise(x, 1e-3, 42, 0)), 2)
        r["quant_bin (1R+1W+stats)"] = gbs(
            timeit(lambda: ops.quantize_dequantize(y, 256, 0.5)), 2)
        results[f"n={n>>20}M floats ({4*n>>20} MiB)"] = {
            k: round(v, 1) for k, v in r.items()}
    print(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()

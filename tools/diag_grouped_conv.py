"""Probe: would a grouped-conv mega round pay for ResNet?

Compares fwd+bwd time of K=10 clients' convs run (a) sequentially as
K separate F.conv2d calls vs (b) as ONE grouped conv (groups=K over
K-stacked channels), for the fed-CIFAR100 ResNet-18 shapes (input
3x24x24, ImageNet stem).  Decision gate: proceed with the mega design
only if grouped wins >=2x end-to-end-ish.
"""

import sys
import time

import torch
import torch.nn.functional as F

K, BS = 10, 20
# (C_in, C_out, H, W, stride, kernel) — representative layer shapes
SHAPES = [
    (3, 64, 24, 24, 2, 7),     # stem
    (64, 64, 6, 6, 1, 3),      # layer1 (24->12 pool->6... spatial approx)
    (64, 128, 6, 6, 2, 3),
    (128, 128, 3, 3, 1, 3),
    (256, 256, 2, 2, 1, 3),
    (512, 512, 1, 1, 1, 3),
]


def bench(fn, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / n * 1e3


def main():
    torch.backends.cudnn.benchmark = True
    tot_seq = tot_grp = 0.0
    for (ci, co, h, w, st, ks) in SHAPES:
        xs = [torch.randn(BS, ci, h, w, device="cuda", requires_grad=True)
              for _ in range(K)]
        ws = [torch.randn(co, ci, ks, ks, device="cuda",
                          requires_grad=True) for _ in range(K)]
        xg = torch.randn(BS, K * ci, h, w, device="cuda",
                         requires_grad=True)
        wg = torch.randn(K * co, ci, ks, ks, device="cuda",
                         requires_grad=True)

        def seq():
            outs = [F.conv2d(x, wk, stride=st, padding=ks // 2)
                    for x, wk in zip(xs, ws)]
            s = sum(o.square().sum() for o in outs)
            s.backward()

        def grp():
            o = F.conv2d(xg, wg, stride=st, padding=ks // 2, groups=K)
            o.square().sum().backward()

        ms_s, ms_g = bench(seq), bench(grp)
        tot_seq += ms_s
        tot_grp += ms_g
        print(f"c{ci:>4}->{co:<4} {h}x{w} s{st} k{ks}:  "
              f"seq {ms_s:7.3f} ms   grouped {ms_g:7.3f} ms   "
              f"ratio {ms_s / ms_g:4.2f}x")
    print(f"TOTAL: seq {tot_seq:.2f} ms  grouped {tot_grp:.2f} ms  "
          f"ratio {tot_seq / tot_grp:.2f}x")


if __name__ == "__main__":
    main()

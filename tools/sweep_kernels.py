#!/usr/bin/env python3
"""Empirical sweeps for the slow NHWC reduction kernels: compares our
kernels against torch-native equivalents at AlexNet shapes, to locate
the gap (launch config vs memory pattern vs kernel structure).
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

DEV = "cuda:0"


def timeit(fn, reps=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3


def main():
    print("device:", torch.cuda.get_device_name(0))
    print("== col_sum: ours vs torch.sum(dim=0) ==")
    for (M, N) in [(774400, 96), (186624, 256), (43264, 384),
                   (256, 4096)]:
        a = torch.randn(M, N, device=DEV).bfloat16()
        out = torch.empty(N, device=DEV, dtype=torch.float32)
        t_ours = timeit(lambda: ops.col_sum(a, out))
        t_torch = timeit(lambda: a.float().sum(dim=0))
        t_torch2 = timeit(lambda: a.sum(dim=0, dtype=torch.float32))
        gb = M * N * 2 / 1e9
        print(f"M={M:7d} N={N:5d}: ours {t_ours:7.1f}us "
              f"({gb/t_ours*1e6:5.0f} GB/s)  torch(f32cast) "
              f"{t_torch:7.1f}us  torch(direct) {t_torch2:7.1f}us "
              f"({gb/t_torch2*1e6:5.0f} GB/s)")

    print("== LRN fwd: ours vs torch composite ==")
    for (rows, C) in [(774400, 96), (186624, 256)]:
        x = torch.randn(rows, C, device=DEV).bfloat16()
        y = torch.empty_like(x)
        denom = torch.empty(rows, C, device=DEV, dtype=torch.float32)

        def torch_lrn():
            xf = x.float()
            sq = xf * xf
            # cross-channel window sum via conv1d-free cumsum trick
            cs = torch.nn.functional.pad(sq, (2, 2)).unfold(
                1, 5, 1).sum(-1)
            d = 2.0 + 1e-4 * cs
            return (xf * d.pow(-0.75)).bfloat16()
        t_ours = timeit(lambda: ops.lrn_fwd(x, y, denom, 1e-4, 0.75,
                                            2.0, 5))
        t_t = timeit(torch_lrn)
        gb = rows * C * 8 / 1e9
        print(f"rows={rows:7d} C={C:4d}: ours {t_ours:7.1f}us "
              f"({gb/t_ours*1e6:5.0f} GB/s)  torch {t_t:7.1f}us")

    print("== pool bwd: ours vs scatter baseline ==")
    for (N, H, W, C) in [(256, 55, 55, 96), (256, 27, 27, 256)]:
        OH = (H - 3) // 2 + 1
        x = torch.randn(N, H, W, C, device=DEV).bfloat16()
        y = torch.empty(N, OH, OH, C, device=DEV, dtype=torch.bfloat16)
        offs = torch.empty(N * OH * OH * C, device=DEV,
                           dtype=torch.int32)
        desc = [N, H, W, C, 3, 3, 2, 2, OH, OH, 0]
        ops.pool_fwd(x.view(N, -1), y.view(N, -1), offs, desc, 0)
        ey = torch.randn_like(y)
        ex = torch.empty_like(x)
        t_ours = timeit(lambda: ops.pool_bwd(ey.view(N, -1),
                                             ex.view(N, -1), offs,
                                             desc, 0))

        def torch_scatter():
            exf = torch.zeros(N * H * W * C, device=DEV,
                              dtype=torch.float32)
            exf.scatter_add_(0, offs.long(),
                             ey.view(-1).float())
            return exf.bfloat16()
        t_t = timeit(torch_scatter)
        gb = (N * H * W * C * 4 + N * OH * OH * C * 6) / 1e9
        print(f"{N}x{H}x{W}x{C}: ours {t_ours:7.1f}us "
              f"({gb/t_ours*1e6:5.0f} GB/s)  torch-scatter {t_t:7.1f}us")

    print("== plain streaming ceiling (read bf16, add, write bf16) ==")
    for n in (74_000_000,):
        a = torch.randn(n, device=DEV).bfloat16()
        t = timeit(lambda: a.add(1.0))
        print(f"n={n}: torch add {t:7.1f}us "
              f"({n*4/1e9/t*1e6:5.0f} GB/s)")
        err = torch.randn(n, device=DEV).bfloat16()
        yb = torch.randn(n, device=DEV).bfloat16()
        t2 = timeit(lambda: ops.act_bwd_inplace(err, yb, 3))
        print(f"n={n}: act_bwd {t2:7.1f}us ({n*6/1e9/t2*1e6:5.0f} GB/s)")


if __name__ == "__main__":
    main()

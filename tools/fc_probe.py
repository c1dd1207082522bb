#!/usr/bin/env python3
"""FC GEMM deep probe (AlexNet batch-256 shapes, MI355X).

The FC block is ~11% of AlexNet kernel time; its GEMMs are small-M
(fwd/dgrad: M=256) or small-K (wgrad: K=256) — the fill-the-chip
regime, not the square-tile one.  This probe measures, per shape:
  * our gemm_nt / gemm_tn at several split-K budgets (set_routing)
  * torch.matmul (hipBLASLt) as the library ceiling reference
so the stream-K / small-M tier decision is data, not guesswork.

    gpurun -- 'python tools/fc_probe.py > gpurun_out/fc_probe.txt 2>&1'
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

DEV = "cuda:0"
REPS = 50


def timeit(fn, reps=REPS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1000.0 / reps   # µs


def tf(us, flops):
    return flops / (us * 1e-6) / 1e12


# (name, M, N, K): fwd/dgrad are gemm_nt A[M,K]·B[N,K]T
NT_SHAPES = [
    ("fc6 fwd", 256, 4096, 9216),
    ("fc7 fwd", 256, 4096, 4096),
    ("fc8 fwd", 256, 1000, 4096),
    ("fc7 dgrad", 256, 4096, 4096),
    ("fc6 dgrad", 256, 9216, 4096),   # ei wide
]
# wgrad gemm_tn: A[K,M], B[K,N] -> C[M,N] fp32, K=batch
TN_SHAPES = [
    ("fc6 wgrad", 4096, 9216, 256),
    ("fc7 wgrad", 4096, 4096, 256),
    ("fc8 wgrad", 1000, 4096, 256),
]


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    print("device:", torch.cuda.get_device_name(0))
    for (nm, M, N, K) in NT_SHAPES:
        a = torch.randn(M, K, device=DEV).bfloat16()
        b = torch.randn(N, K, device=DEV).bfloat16()
        c = torch.empty(M, N, device=DEV, dtype=torch.bfloat16)
        bias = torch.randn(N, device=DEV)
        flops = 2.0 * M * N * K
        for budget in (128, 256, 512, 1024, 2048):
            ops.require().set_routing(budget, -1, -1, -1)
            us = timeit(lambda: ops.gemm_nt(a, b, c, bias, 3))
            print("%-12s M%5d N%5d K%5d nt_budget %4d: %7.1f us %6.1f TF"
                  % (nm, M, N, K, budget, us, tf(us, flops)))
        ops.require().set_routing(512, -1, -1, -1)
        # library ceiling (no bias/act fusion)
        us = timeit(lambda: torch.matmul(a, b.t()))
        print("%-12s %40s: %7.1f us %6.1f TF  (hipBLASLt)"
              % (nm, "torch.matmul", us, tf(us, flops)))
    for (nm, M, N, K) in TN_SHAPES:
        at = torch.randn(K, M, device=DEV).bfloat16()
        bt = torch.randn(K, N, device=DEV).bfloat16()
        ct = torch.zeros(M, N, device=DEV, dtype=torch.float32)
        flops = 2.0 * M * N * K
        for budget in (256, 1024, 4096, 16384):
            ops.require().set_routing(-1, budget, budget, -1)
            us = timeit(lambda: ops.gemm_tn(at, bt, ct))
            print("%-12s M%5d N%5d K%5d tn_budget %5d: %7.1f us %6.1f TF"
                  % (nm, M, N, K, budget, us, tf(us, flops)))
        ops.require().set_routing(-1, 1024, 1024, -1)
        us = timeit(lambda: torch.matmul(at.t().float(), bt.float()))
        print("%-12s %40s: %7.1f us %6.1f TF  (f32 matmul ref)"
              % (nm, "torch.matmul", us, tf(us, flops)))
        us = timeit(lambda: torch.matmul(at.t(), bt))
        print("%-12s %40s: %7.1f us %6.1f TF  (bf16 matmul)"
              % (nm, "torch.matmul bf16", us, tf(us, flops)))


if __name__ == "__main__":
    main()

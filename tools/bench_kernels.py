#!/usr/bin/env python3
"""Per-kernel microbenchmark at AlexNet batch-256 shapes (MI355X).

Times each hot _mzk op in isolation with hipEvents and prints
achieved GB/s and TFLOP/s next to a roofline estimate, so kernel work
can be targeted where the machine is actually idle.

    gpurun -- 'python tools/bench_kernels.py > gpurun_out/kbench.txt'
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

DEV = "cuda:0"
REPS = 20

# AlexNet conv shapes at batch 256, C padded to 8 (input) as the loader
# does:  (name, N, H, W, C, K, KH, KW, SY, SX, PT, PL)
CONVS = [
    ("conv1", 256, 227, 227, 8, 96, 11, 11, 4, 4, 0, 0),
    ("conv1c4", 256, 227, 227, 4, 96, 11, 11, 4, 4, 0, 0),
    ("conv2", 256, 27, 27, 96, 256, 5, 5, 1, 1, 2, 2),
    ("conv3", 256, 13, 13, 256, 384, 3, 3, 1, 1, 1, 1),
    ("conv4", 256, 13, 13, 384, 384, 3, 3, 1, 1, 1, 1),
    ("conv5", 256, 13, 13, 384, 256, 3, 3, 1, 1, 1, 1),
]

POOLS = [  # (name, N, H, W, C, k, s)
    ("pool1", 256, 55, 55, 96, 3, 2),
    ("pool2", 256, 27, 27, 256, 3, 2),
    ("pool5", 256, 13, 13, 256, 3, 2),
]

LRNS = [  # (name, N, H, W, C)
    ("lrn1", 256, 55, 55, 96),
    ("lrn2", 256, 27, 27, 256),
]

FCS = [  # (name, M, N, K)
    ("fc6", 256, 4096, 9216),
    ("fc7", 256, 4096, 4096),
    ("fc8", 256, 1000, 4096),
]


def timeit(fn, reps=REPS):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3  # us


def report(name, us, flops=0, bytes_=0):
    tf = flops / us / 1e6 if flops else 0
    gb = bytes_ / us / 1e3 if bytes_ else 0
    print(f"{name:28s} {us:9.1f} us   {tf:8.1f} TF/s   {gb:8.0f} GB/s")


def out_hw(h, k, s, p0, p1):
    return (h + p0 + p1 - k) // s + 1


def bench_convs():
    print("== conv fwd / dgrad / wgrad (bf16, MFMA) ==")
    for (nm, N, H, W, C, K, KH, KW, SY, SX, PT, PL) in CONVS:
        OH = out_hw(H, KH, SY, PT, PT)
        OW = out_hw(W, KW, SX, PL, PL)
        desc = [N, H, W, C, K, KH, KW, OH, OW, SY, SX, PT, PL]
        x = torch.randn(N, H, W, C, device=DEV).bfloat16()
        w = torch.randn(K, KH * KW * C, device=DEV).bfloat16()
        wd = torch.randn(C, KH, KW, K, device=DEV).bfloat16().contiguous()
        bias = torch.randn(K, device=DEV)
        y = torch.empty(N, OH, OW, K, device=DEV, dtype=torch.bfloat16)
        eo = torch.randn(N, OH, OW, K, device=DEV).bfloat16()
        ei = torch.empty(N, H, W, C, device=DEV, dtype=torch.bfloat16)
        gw = torch.zeros(K, KH * KW * C, device=DEV, dtype=torch.float32)
        flops = 2.0 * N * OH * OW * K * KH * KW * C
        us = timeit(lambda: ops.conv_fwd(
            x.view(N, -1), w, y.view(N * OH * OW, K), bias, desc, 3))
        report(nm + " fwd", us, flops)
        wdg = wd.view(C, KH * KW * K)
        us = timeit(lambda: ops.conv_dgrad(
            eo.view(N * OH * OW, K), wdg, ei.view(N * H * W, C), desc))
        report(nm + " dgrad", us, flops)
        us = timeit(lambda: ops.conv_wgrad(
            eo.view(N * OH * OW, K), x.view(N, -1), gw, desc, 0.0))
        report(nm + " wgrad", us, flops)
        bsum = torch.empty(K, device=DEV, dtype=torch.float32)
        us = timeit(lambda: ops.col_sum(eo.view(N * OH * OW, K), bsum))
        report(nm + " bias(col_sum)", us, 0,
               N * OH * OW * K * 2)


def bench_pools():
    print("== pooling (max) ==")
    for (nm, N, H, W, C, k, s) in POOLS:
        OH = (H - k + s - 1) // s + 1  # ceil mode off: floor
        OH = (H - k) // s + 1
        OW = OH
        x = torch.randn(N, H, W, C, device=DEV).bfloat16()
        y = torch.empty(N, OH, OW, C, device=DEV, dtype=torch.bfloat16)
        offs = torch.empty(N * OH * OW * C, device=DEV, dtype=torch.uint8)
        ey = torch.randn(N, OH, OW, C, device=DEV).bfloat16()
        ex = torch.empty_like(x)
        desc = [N, H, W, C, k, k, s, s, OH, OW, 0]
        us = timeit(lambda: ops.pool_fwd(x.view(N, -1), y.view(N, -1),
                                         offs, desc, 0))
        report(nm + " fwd", us, 0, (N * H * W * C + N * OH * OW * C) * 2)
        us = timeit(lambda: ops.pool_bwd(ey.view(N, -1), ex.view(N, -1),
                                         offs, desc, 0))
        report(nm + " bwd", us, 0, (N * H * W * C * 2 + N * OH * OW * C *
                                    6))


def bench_lrns():
    print("== LRN (win=5) ==")
    for (nm, N, H, W, C) in LRNS:
        rows = N * H * W
        x = torch.randn(N, H, W, C, device=DEV).bfloat16()
        y = torch.empty_like(x)
        denom = torch.empty(N, H, W, C, device=DEV, dtype=torch.float32)
        ey = torch.randn_like(x)
        ex = torch.empty_like(x)
        nb = rows * C
        # NOTE: pass 4-D tensors — the binding reads C from size(-1)
        us = timeit(lambda: ops.lrn_fwd(x, y, denom, 1e-4, 0.75, 2.0, 5))
        report(nm + " fwd", us, 0, nb * (2 + 2 + 4))
        us = timeit(lambda: ops.lrn_bwd(ey, x, None, denom, ex,
                                        1e-4, 0.75, 2.0, 5))
        report(nm + " bwd", us, 0, nb * (2 * 3 + 4 * 3 + 2))


def bench_fcs():
    print("== FC GEMMs (bf16 MFMA) ==")
    for (nm, M, N, K) in FCS:
        a = torch.randn(M, K, device=DEV).bfloat16()
        b = torch.randn(N, K, device=DEV).bfloat16()
        c = torch.empty(M, N, device=DEV, dtype=torch.bfloat16)
        bias = torch.randn(N, device=DEV)
        flops = 2.0 * M * N * K
        us = timeit(lambda: ops.gemm_nt(a, b, c, bias, 3))
        report(nm + " fwd(nt)", us, flops)
        at = torch.randn(K, M, device=DEV).bfloat16()
        bt = torch.randn(K, N, device=DEV).bfloat16()
        ct = torch.empty(M, N, device=DEV, dtype=torch.float32)
        us = timeit(lambda: ops.gemm_tn(at, bt, ct))
        report(nm + " wgrad(tn)", us, flops)


def bench_misc():
    print("== elementwise ==")
    n = 256 * 4096
    err = torch.randn(n, device=DEV).bfloat16()
    yb = torch.randn(n, device=DEV).bfloat16()
    us = timeit(lambda: ops.act_bwd_inplace(err, yb, 3))
    report("act_bwd 1M", us, 0, n * 6)
    for pn in (61_000_000, 4_000_000):
        w = torch.randn(pn // 16, device=DEV, dtype=torch.float32)
        g = torch.randn_like(w)
        v = torch.zeros_like(w)

        class FA:
            devmem = w
            bf16 = w.bfloat16()
        us = timeit(lambda: ops.sgd_step(FA, g, v, 0.01, 1e-4, 0.0, 0.9,
                                         1 / 256))
        report("sgd %dK" % (pn // 16000), us, 0, (pn // 16) * 20)


if __name__ == "__main__":
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    print("device:", torch.cuda.get_device_name(0))
    bench_convs()
    bench_pools()
    bench_lrns()
    bench_fcs()
    bench_misc()

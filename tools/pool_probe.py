#!/usr/bin/env python3
"""Per-shape pooling fwd/bwd probe (AlexNet b256 pools).

pool_bwd is 5.1% of the AlexNet step at ~3x its streaming floor; this
probe times each pool shape separately (the kernel table only shows
the 3-shape average) and is the rocprofv3 --pmc target for deciding
whether the window-search backward is memory-wait or issue bound.

    gpurun -- 'python tools/pool_probe.py; rocprofv3 --pmc SQ_WAVE_CYCLES \
       SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_INSTS_VALU \
       --kernel-trace -d gpurun_out/pmc_pool -- python tools/pool_probe.py'
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

DEV = "cuda:0"
POOLS = [  # (name, N, H, W, C, k, s)
    ("pool1", 256, 55, 55, 96, 3, 2),
    ("pool2", 256, 27, 27, 256, 3, 2),
    ("pool5", 256, 13, 13, 256, 3, 2),
]


def timeit(fn, reps=30):
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
    return s.elapsed_time(e) * 1000.0 / reps


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    for (nm, N, H, W, C, k, s) in POOLS:
        OH = (H - k + s - 1) // s + 1
        OW = (W - k + s - 1) // s + 1
        desc = [N, H, W, C, k, k, s, s, OH, OW, 0]
        x = torch.randn(N, H, W, C, device=DEV).bfloat16()
        y = torch.empty(N, OH, OW, C, device=DEV, dtype=torch.bfloat16)
        offs = torch.empty(N * OH * OW * C, device=DEV,
                           dtype=torch.uint8)
        ey = torch.randn(N, OH, OW, C, device=DEV).bfloat16()
        ex = torch.empty_like(x)
        ops.pool_fwd(x.view(-1, C), y.view(-1, C), offs, desc, 0)
        us_f = timeit(lambda: ops.pool_fwd(x.view(-1, C), y.view(-1, C),
                                           offs, desc, 0))
        us_b = timeit(lambda: ops.pool_bwd(ey.view(-1, C),
                                           ex.view(-1, C), offs, desc,
                                           0))
        in_mb = N * H * W * C * 2 / 1e6
        out_mb = N * OH * OW * C * 2 / 1e6
        floor_b = (in_mb + out_mb + out_mb / 2) / 8e3 * 1e3  # us @8TB/s
        print("%s  fwd %7.1f us   bwd %7.1f us  (bwd stream floor "
              "~%5.1f us; ex %d MB ey %d MB)"
              % (nm, us_f, us_b, floor_b, int(in_mb), int(out_mb)))


if __name__ == "__main__":
    main()

#!/usr/bin/env python3
"""Sweep per-shape split-K for the dense (FC) GEMM entry points and
emit the ``fc_nt_splits`` / ``fc_tn_splits`` rows of
gfx950_routing.json — the autotune DB one level deeper than the global
budgets (measured: the budget formula over-splits small-tile FC shapes;
fc8 wgrad 33.1→12.9 µs, fc8 fwd 26.0→19.0 µs at batch 256).

Run on an MI355X:
    gpurun -- 'python tools/gen_fc_splits.py > gpurun_out/fc_splits.json'
then merge the JSON into veles/znicz_amd/ops/gfx950_routing.json and
commit.  A row is only emitted when the best split beats the current
formula by >3% (no-regression guard); splits<=0 rows are never written.
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(
    os.path.abspath(__file__)), ".."))

import torch  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

DEV = "cuda:0"
REPS = 50

# the model zoo's dense-GEMM shapes (AlexNet b256, VGG-16 b64/b256):
# NT = fwd/dgrad A[M,K]·B[N,K]T; TN = wgrad A[K,M]T·B[K,N]
NT_SHAPES = [
    (256, 4096, 9216), (256, 4096, 4096), (256, 1000, 4096),
    (256, 9216, 4096), (256, 4096, 1000),
    (64, 4096, 25088), (64, 4096, 4096), (64, 1000, 4096),
    (64, 25088, 4096), (64, 4096, 1000),
    (256, 4096, 25088), (256, 25088, 4096),
]
TN_SHAPES = [
    (4096, 9216, 256), (4096, 4096, 256), (1000, 4096, 256),
    (4096, 25088, 64), (4096, 4096, 64), (1000, 4096, 64),
    (4096, 25088, 256),
]
CANDS = [1, 2, 3, 4, 6, 8, 12, 16, 24, 32, 48, 64]


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
    return s.elapsed_time(e) * 1000.0 / reps


def sweep(is_tn, M, N, K, run):
    mod = ops.require()
    mod.set_fc_split(is_tn, M, N, K, -1)      # formula baseline
    base = min(timeit(run), timeit(run))
    best_s, best_t = 0, base
    nk = (K + 63) // 64 if not is_tn else (K + 31) // 32
    for s in CANDS:
        if s > nk:
            break
        mod.set_fc_split(is_tn, M, N, K, s)
        t = timeit(run)
        if t < best_t:
            best_s, best_t = s, t
    mod.set_fc_split(is_tn, M, N, K, -1)
    keep = best_s > 0 and best_t < base * 0.97
    sys.stderr.write(
        "%s M%-6d N%-6d K%-6d formula %7.1f us -> best s=%-2d %7.1f us"
        " %s\n" % ("tn" if is_tn else "nt", M, N, K, base, best_s,
                   best_t, "KEEP" if keep else "(within guard)"))
    return best_s if keep else 0


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    out = {"fc_nt_splits": [], "fc_tn_splits": []}
    for (M, N, K) in NT_SHAPES:
        a = torch.randn(M, K, device=DEV).bfloat16()
        b = torch.randn(N, K, device=DEV).bfloat16()
        c = torch.empty(M, N, device=DEV, dtype=torch.bfloat16)
        bias = torch.randn(N, device=DEV)
        s = sweep(0, M, N, K, lambda: ops.gemm_nt(a, b, c, bias, 3))
        if s:
            out["fc_nt_splits"].append([M, N, K, s])
        del a, b, c, bias
    for (M, N, K) in TN_SHAPES:
        at = torch.randn(K, M, device=DEV).bfloat16()
        bt = torch.randn(K, N, device=DEV).bfloat16()
        ct = torch.zeros(M, N, device=DEV, dtype=torch.float32)
        s = sweep(1, M, N, K, lambda: ops.gemm_tn(at, bt, ct))
        if s:
            out["fc_tn_splits"].append([M, N, K, s])
        del at, bt, ct
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()

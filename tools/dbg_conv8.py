#!/usr/bin/env python3
"""Validate the 8-phase conv fwd/dgrad path against torch.conv2d on
shapes large enough to route to gemm_nt8_k (M >= 2048, N >= 192)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import numpy
import torch
import torch.nn.functional as F

from veles.znicz_amd import ops

RNG = numpy.random.RandomState(5)


def check(name, got, want, tol=0.06):
    g = got.float().cpu().numpy()
    w = want.float().cpu().numpy()
    err = numpy.abs(g - w)
    lim = tol + 0.03 * numpy.abs(w)
    bad = err > lim
    mx = float(err.max())
    print(f"{name}: bad {bad.sum()}/{bad.size} maxerr {mx:.4f}")
    return bad.sum() == 0


def run_case(N, H, W, C, K, KH, SY, PT):
    KW, SX, PL = KH, SY, PT
    OH = (H + 2 * PT - KH) // SY + 1
    OW = (W + 2 * PL - KW) // SX + 1
    desc = [N, H, W, C, K, KH, KW, OH, OW, SY, SX, PT, PL]
    x = torch.randn(N, H, W, C, device="cuda").bfloat16()
    w = torch.randn(K, KH, KW, C, device="cuda").bfloat16() * 0.2
    bias = torch.randn(K, device="cuda")
    y = torch.empty(N, OH, OW, K, device="cuda", dtype=torch.bfloat16)
    ops.conv_fwd(x.view(N, -1), w.view(K, -1).contiguous(),
                 y.view(-1, K), bias, desc, 0)
    # torch reference (fp32)
    xf = x.float().permute(0, 3, 1, 2)
    wf = w.float().permute(0, 3, 1, 2)
    ref = F.conv2d(xf, wf, bias=bias, stride=SY, padding=PT)
    ok1 = check(f"fwd {N}x{H}x{W}x{C}->K{K} k{KH}s{SY}",
                y.permute(0, 3, 1, 2).float(), ref)

    eo = (torch.randn(N, OH, OW, K, device="cuda") * 0.2).bfloat16()
    wdg = w.permute(3, 1, 2, 0).contiguous()     # [C,KH,KW,K]
    ei = torch.empty(N, H, W, C, device="cuda", dtype=torch.bfloat16)
    ops.conv_dgrad(eo.view(-1, K), wdg.view(C, -1), ei.view(-1, C),
                   desc)
    ref_ei = torch.nn.grad.conv2d_input(
        (N, C, H, W), wf, eo.float().permute(0, 3, 1, 2), stride=SY,
        padding=PT)
    ok2 = check(f"dgrad same", ei.permute(0, 3, 1, 2).float(), ref_ei)
    return ok1 and ok2


def main():
    allok = True
    # big shapes -> nt8 route; includes non-multiple-of-256 M and N
    allok &= run_case(32, 27, 27, 96, 256, 5, 1, 2)     # conv2-like
    allok &= run_case(32, 13, 13, 256, 384, 3, 1, 1)    # conv3-like
    allok &= run_case(16, 13, 13, 384, 384, 3, 1, 1)    # conv4-like
    allok &= run_case(8, 30, 30, 64, 200, 3, 1, 1)      # odd N=200
    allok &= run_case(4, 57, 57, 128, 512, 3, 1, 1)     # vgg-like
    print("ALL OK" if allok else "FAILURES")
    sys.exit(0 if allok else 1)


if __name__ == "__main__":
    main()

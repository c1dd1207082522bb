"""GEMM-core shape matrix vs torch references (MI355X).

Covers the routing space of the three cores: 128² NT (+split-K FC
path), TN2 (tr_b16 wgrad), and — when VELES_NT8 is set — the 8-phase
256² core.  These are the shapes the probes (tools/dbg_tn2.py,
tools/dbg_conv8.py) used to find the split-K / conditional-load /
tr_b16-race regressions; kept as tests so the routes stay guarded.
"""

import numpy
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

import torch.nn.functional as F  # noqa: E402

from veles.znicz_amd import ops  # noqa: E402

RNG = numpy.random.RandomState(29)


def dev_bf16(a):
    return torch.from_numpy(numpy.ascontiguousarray(a)).to(
        "cuda", torch.bfloat16)


@pytest.mark.parametrize("M,N,K", [
    (64, 64, 32), (64, 64, 64), (64, 64, 192), (256, 128, 512),
    (1024, 2048, 64), (100, 72, 96),
])
def test_gemm_tn_matrix(M, N, K):
    a = RNG.uniform(-1, 1, (K, M)).astype(numpy.float32)
    b = RNG.uniform(-1, 1, (K, N)).astype(numpy.float32)
    out = torch.zeros((M, N), dtype=torch.float32, device="cuda")
    ops.gemm_tn(dev_bf16(a), dev_bf16(b), out)
    want = dev_bf16(a).float().cpu().numpy().T.dot(
        dev_bf16(b).float().cpu().numpy())
    numpy.testing.assert_allclose(
        out.cpu().numpy(), want, rtol=2e-2,
        atol=0.05 * max(1.0, K / 500))


@pytest.mark.parametrize("M,N,K", [
    (256, 4096, 9216),   # fc6: split-K route
    (96, 200, 128),      # small with bias path
    (256, 1000, 4096),   # fc8
    (96, 48, 64),        # narrow-N BNT=64 route, tail cols
    (64, 20, 64),        # narrow-N BNT=32 route, tail cols
    (300, 64, 256),      # narrow-N BNT=64 exact, M tail
])
def test_gemm_nt_fc_matrix(M, N, K):
    a = RNG.uniform(-1, 1, (M, K)).astype(numpy.float32)
    b = RNG.uniform(-1, 1, (N, K)).astype(numpy.float32)
    bias = RNG.uniform(-1, 1, N).astype(numpy.float32)
    out = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    ops.gemm_nt(dev_bf16(a), dev_bf16(b), out,
                bias=torch.from_numpy(bias).cuda(), activation=3)
    want = numpy.maximum(
        dev_bf16(a).float().cpu().numpy().dot(
            dev_bf16(b).float().cpu().numpy().T) + bias, 0.0)
    numpy.testing.assert_allclose(
        out.float().cpu().numpy(), want, rtol=2e-2,
        atol=0.06 * max(1.0, K / 500))


@pytest.mark.parametrize("N,H,W,C,K,KH,SY,PT", [
    (32, 27, 27, 96, 256, 5, 1, 2),
    (16, 13, 13, 384, 384, 3, 1, 1),
    (8, 30, 30, 64, 200, 3, 1, 1),
    (4, 23, 23, 8, 96, 11, 4, 0),      # conv1-like strided
    (16, 32, 32, 8, 32, 5, 1, 2),      # cifar conv1: narrow-N fwd
    (8, 16, 16, 32, 32, 5, 1, 2),      # cifar conv2: narrow fwd+dgrad
    (8, 8, 8, 32, 64, 5, 1, 2),        # cifar conv3: BNT=64 fwd
    (4, 23, 23, 4, 96, 11, 4, 0),      # alexnet conv1: C=4 packed DMA
    (8, 15, 15, 4, 32, 3, 1, 1),       # C=4 WITH padding: all-slow path
    (24, 19, 19, 96, 96, 5, 1, 2),     # deep ntn8 BNT=96 fwd+dgrad,
                                       # ragged M (24*361 % 256 != 0)
    (20, 21, 21, 64, 72, 3, 1, 1),     # deep ntn8 BNT=64 dgrad (M<2^20,
                                       # K=648>=512), ragged tiles
    (6, 17, 17, 16, 48, 5, 2, 2),      # residue-class strided dgrad
                                       # WITH padding (deconv-AE shape)
])
def test_conv_fwd_dgrad_vs_torch(N, H, W, C, K, KH, SY, PT):
    KW, SX, PL = KH, SY, PT
    OH = (H + 2 * PT - KH) // SY + 1
    OW = (W + 2 * PL - KW) // SX + 1
    desc = [N, H, W, C, K, KH, KW, OH, OW, SY, SX, PT, PL]
    x = (torch.randn(N, H, W, C, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(K, KH, KW, C, device="cuda") * 0.2).bfloat16()
    bias = torch.randn(K, device="cuda")
    y = torch.empty(N, OH, OW, K, device="cuda", dtype=torch.bfloat16)
    ops.conv_fwd(x.view(N, -1), w.view(K, -1).contiguous(),
                 y.view(-1, K), bias, desc, 0)
    xf = x.float().permute(0, 3, 1, 2)
    wf = w.float().permute(0, 3, 1, 2)
    ref = F.conv2d(xf, wf, bias=bias, stride=SY, padding=PT)
    numpy.testing.assert_allclose(
        y.permute(0, 3, 1, 2).float().cpu().numpy(),
        ref.cpu().numpy(), rtol=3e-2, atol=0.15)

    eo = (torch.randn(N, OH, OW, K, device="cuda") * 0.2).bfloat16()
    wdg = w.permute(3, 1, 2, 0).contiguous()
    ei = torch.empty(N, H, W, C, device="cuda", dtype=torch.bfloat16)
    ops.conv_dgrad(eo.view(-1, K), wdg.view(C, -1), ei.view(-1, C),
                   desc)
    ref_ei = torch.nn.grad.conv2d_input(
        (N, C, H, W), wf, eo.float().permute(0, 3, 1, 2), stride=SY,
        padding=PT)
    numpy.testing.assert_allclose(
        ei.permute(0, 3, 1, 2).float().cpu().numpy(),
        ref_ei.cpu().numpy(), rtol=3e-2, atol=0.1)

    gw = torch.zeros(K, KH * KW * C, device="cuda",
                     dtype=torch.float32)
    gb = torch.full((K,), 7.0, device="cuda", dtype=torch.float32)
    fused = ops.conv_wgrad(eo.view(-1, K), x.view(N, -1), gw, desc, 0.0,
                           bias_out=gb)
    ref_gw = torch.nn.grad.conv2d_weight(
        xf, (K, C, KH, KW), eo.float().permute(0, 3, 1, 2), stride=SY,
        padding=PT)
    numpy.testing.assert_allclose(
        gw.view(K, KH, KW, C).permute(0, 3, 1, 2).cpu().numpy(),
        ref_gw.cpu().numpy(), rtol=3e-2,
        atol=0.3 + 0.001 * N * OH * OW)
    # fused bias gradient = column sums of err_output (beta=0 path);
    # fusion is narrow-tile only (M=K<=64 on the vectorized TN2 route,
    # which accepts any C%4==0 via the packed (kw,c) staging)
    assert fused == ((K % 8 == 0) and (C % 4 == 0) and K <= 64)
    if fused:
        ref_gb = eo.float().view(-1, K).sum(0)
        numpy.testing.assert_allclose(
            gb.cpu().numpy(), ref_gb.cpu().numpy(), rtol=2e-2,
            atol=0.1 + 0.0005 * N * OH * OW)

import sys, os
sys.path.insert(0, "/root/repo")
import torch, numpy
from veles.znicz_amd import ops
RNG = numpy.random.RandomState(1)
def dev_bf16(a): return torch.from_numpy(numpy.ascontiguousarray(a)).to("cuda", torch.bfloat16)
def bfr(a): return dev_bf16(a).float().cpu().numpy()
for (M,N,K) in [(64,64,32),(64,64,64),(64,64,96),(64,64,128),(64,64,160),(64,64,192),(64,64,256),(1024,16384,64),(256,128,512)]:
    a = RNG.uniform(-1,1,(K,M)).astype(numpy.float32)
    b = RNG.uniform(-1,1,(K,N)).astype(numpy.float32)
    out = torch.zeros((M,N), dtype=torch.float32, device="cuda")
    ops.gemm_tn(dev_bf16(a), dev_bf16(b), out)
    want = bfr(a).T.dot(bfr(b))
    got = out.cpu().numpy()
    err = numpy.abs(got-want)
    bad = err > (0.05*max(1,K/500) + 2e-2*numpy.abs(want))
    print(f"M{M} N{N} K{K}: bad {bad.sum()}/{bad.size}", end="")
    if bad.sum():
        bi = numpy.argwhere(bad)
        print("  first bad:", bi[:4].tolist(), " rows with bad:", sorted(set(bi[:,0].tolist()))[:10], " cols:", sorted(set(bi[:,1].tolist()))[:10], end="")
        r,c = bi[0]
        print(f"  got={got[r,c]:.3f} want={want[r,c]:.3f}")
    else:
        print()

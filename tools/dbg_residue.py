import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import numpy, torch
import torch.nn.functional as F
from veles.znicz_amd import ops
torch.manual_seed(2)
N,H,W,C,K,KH,SY,PT = 4,23,23,8,96,11,4,0
KW,SX,PL = KH,SY,PT
OH=(H+2*PT-KH)//SY+1; OW=(W+2*PL-KW)//SX+1
desc=[N,H,W,C,K,KH,KW,OH,OW,SY,SX,PT,PL]
w = (torch.randn(K,KH,KW,C,device="cuda")*0.2).bfloat16()
eo = (torch.randn(N,OH,OW,K,device="cuda")*0.2).bfloat16()
wdg = w.permute(3,1,2,0).contiguous()
ei = torch.empty(N,H,W,C,device="cuda",dtype=torch.bfloat16)
ops.conv_dgrad(eo.view(-1,K), wdg.view(C,-1), ei.view(-1,C), desc)
wf = w.float().permute(0,3,1,2)
ref = torch.nn.grad.conv2d_input((N,C,H,W), wf, eo.float().permute(0,3,1,2), stride=SY, padding=PT)
got = ei.permute(0,3,1,2).float().cpu().numpy()
want = ref.cpu().numpy()
bad = numpy.abs(got-want) > (0.1 + 3e-2*numpy.abs(want))
print("total bad", bad.sum(), "/", bad.size)
for a in range(SY):
    for b in range(SX):
        m = bad[:, :, a::SY, b::SX]
        print("class a=%d b=%d bad %d/%d" % (a, b, m.sum(), m.size))
# where inside a bad class?
idx = numpy.argwhere(bad)
if len(idx):
    print("ih range of bad:", idx[:,2].min(), idx[:,2].max(),
          "iw:", idx[:,3].min(), idx[:,3].max(),
          "n:", sorted(set(idx[:,0])))

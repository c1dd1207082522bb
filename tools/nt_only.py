import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from veles.znicz_amd import ops
# conv2 fwd + dgrad shapes (the NT kernel's two hot uses)
N,H,W,C,K,KH,KW,SY,SX,PT,PL = 256,27,27,96,256,5,5,1,1,2,2
OH=(H+2*PT-KH)//SY+1; OW=(W+2*PL-KW)//SX+1
desc=[N,H,W,C,K,KH,KW,OH,OW,SY,SX,PT,PL]
x=torch.randn(N,H,W,C,device="cuda").bfloat16()
w=torch.randn(K,KH*KW*C,device="cuda").bfloat16()
wd=torch.randn(C,KH*KW*K,device="cuda").bfloat16()
bias=torch.randn(K,device="cuda")
y=torch.empty(N,OH,OW,K,device="cuda",dtype=torch.bfloat16)
eo=torch.randn(N,OH,OW,K,device="cuda").bfloat16()
ei=torch.empty(N,H,W,C,device="cuda",dtype=torch.bfloat16)
for _ in range(20):
    ops.conv_fwd(x.view(N,-1), w, y.view(-1,K), bias, desc, 3)
    ops.conv_dgrad(eo.view(-1,K), wd, ei.view(-1,C), desc)
torch.cuda.synchronize()
print("done")

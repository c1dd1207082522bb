import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from veles.znicz_amd import ops
N,H,W,C,K,KH,KW,SY,SX,PT,PL = 256,227,227,8,96,11,11,4,4,0,0
OH=(H+2*PT-KH)//SY+1; OW=(W+2*PL-KW)//SX+1
desc=[N,H,W,C,K,KH,KW,OH,OW,SY,SX,PT,PL]
x=torch.randn(N,H,W,C,device="cuda").bfloat16()
w=torch.randn(K,KH*KW*C,device="cuda").bfloat16()
bias=torch.randn(K,device="cuda")
y=torch.empty(N,OH,OW,K,device="cuda",dtype=torch.bfloat16)
eo=torch.randn(N,OH,OW,K,device="cuda").bfloat16()
gw=torch.zeros(K,KH*KW*C,device="cuda",dtype=torch.float32)
for _ in range(20):
    ops.conv_fwd(x.view(N,-1), w, y.view(-1,K), bias, desc, 3)
    ops.conv_wgrad(eo.view(-1,K), x.view(N,-1), gw, desc, 0.0)
torch.cuda.synchronize(); print("done")

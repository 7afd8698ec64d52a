import sys; sys.path.insert(0, "/root/repo")
import torch, math
from bloombee_amd import ops
DEV="cuda:0"
B,Hq,Hkv,ctx,D,P = 32,32,8,2048,128,16
maxp=(ctx+P-1)//P; npages=B*maxp+1
kp=torch.randn(npages,Hkv,P,D,dtype=torch.bfloat16,device=DEV)
vp=torch.randn(npages,Hkv,D,P,dtype=torch.bfloat16,device=DEV)
pt=torch.arange(B*maxp,dtype=torch.int32,device=DEV).reshape(B,maxp)
q=torch.randn(B,Hq,1,D,dtype=torch.bfloat16,device=DEV)*0.1
cl=torch.full((B,),ctx,dtype=torch.int32,device=DEV)
for _ in range(100):
    ops.attn_decode(q,kp,vp,pt,cl,n_split=2)
torch.cuda.synchronize()
print("done")

import math, torch, sys
sys.path.insert(0, "/root/repo")
from colossalai_amd.ops import kernels
_C = kernels()
B,S,Hq,Hkv,D = 8,4096,32,32,128
scale = 1.0/math.sqrt(D)
q = torch.randn(B,S,Hq,D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B,S,Hkv,D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B,S,Hkv,D, device="cuda", dtype=torch.bfloat16)
dout = torch.randn(B,S,Hq,D, device="cuda", dtype=torch.bfloat16)
out, lse = _C.flash_attn_fwd(q,k,v,True,scale)
e = torch.empty(0, device="cuda", dtype=torch.bfloat16)
for _ in range(10):
    _C.flash_attn_fwd(q,k,v,True,scale)
    _C.flash_attn_bwd(dout,q,k,v,out,lse,True,scale,e.clone(),e.clone(),e.clone())
torch.cuda.synchronize()

import torch, math, sys
sys.path.insert(0,"/root/repo")
import metis_amd._hip_ops as ext
B,H,S,D=16,32,2048,128
q=torch.randn(B,H,S,D,device="cuda",dtype=torch.bfloat16)
k=torch.randn_like(q); v=torch.randn_like(q)
sc=1/math.sqrt(D)
for _ in range(3): ext.attn_fwd(q,k,v,sc)
torch.cuda.synchronize()
for _ in range(5): ext.attn_fwd(q,k,v,sc)
torch.cuda.synchronize()

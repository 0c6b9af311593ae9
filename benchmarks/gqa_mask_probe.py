import time, torch
import torch.nn.functional as F
from torch.nn.attention import SDPBackend, sdpa_kernel
DEV="cuda:0"; B,HQ,HKV,D=8,32,8,128; P,Ls=4096,2048; L=P+Ls
q=torch.randn(B,HQ,Ls,D,dtype=torch.bfloat16,device=DEV)
k=torch.randn(B,HKV,L,D,dtype=torch.bfloat16,device=DEV)
v=torch.randn_like(k)
bm=(torch.arange(L,device=DEV)[None,:]<=torch.arange(P,L,device=DEV)[:,None])
am=torch.zeros(Ls,L,dtype=torch.bfloat16,device=DEV).masked_fill_(~bm,float("-inf"))
kr=k.repeat_interleave(HQ//HKV,dim=1); vr=v.repeat_interleave(HQ//HKV,dim=1)
for label,mask,bk in [("rep-kv bool default",bm,None),
                      ("rep-kv add default",am,None),
                      ("rep-kv add EFFICIENT",am,[SDPBackend.EFFICIENT_ATTENTION])]:
    try:
        def c(): return F.scaled_dot_product_attention(q,kr,vr,attn_mask=mask)
        ctx=sdpa_kernel(bk) if bk else None
        if ctx: ctx.__enter__()
        c(); torch.cuda.synchronize(); t0=time.monotonic()
        for _ in range(5): c()
        torch.cuda.synchronize()
        if ctx: ctx.__exit__(None,None,None)
        print(f"{label:24s} {(time.monotonic()-t0)/5*1e3:8.2f} ms",flush=True)
    except Exception as e: print(f"{label:24s} FAIL {str(e)[:60]}",flush=True)

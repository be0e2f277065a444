import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kllms_amd import ops
B,V,K = 8,8192,7
g = torch.Generator(device="cpu").manual_seed(5)
base = torch.randn(B,V,generator=g)*0.01
logits = base.cuda()
t = torch.full((B,),0.9,device="cuda")
tp = torch.ones(B,device="cuda")
tk = torch.full((B,),K,dtype=torch.int32,device="cuda")
seeds = torch.arange(B,dtype=torch.int64,device="cuda")+3
bad=0
for s in range(6):
    steps = torch.full((B,),s,dtype=torch.int64,device="cuda")
    toks,_ = ops.sample(logits,t,tp,tk,seeds,steps)
    for b in range(B):
        sv, si = base[b].sort(descending=True)
        tok = int(toks[b])
        rank = int((si==tok).nonzero()[0])
        if rank >= K:
            bad+=1
            print(f"s={s} b={b} tok={tok} RANK={rank} val={base[b,tok]:.7f} "
                  f"kth={sv[K-1]:.7f} k1th={sv[K]:.7f} gap_kth={float(sv[K-1]-base[b,tok]):.2e}")
print("bad draws:", bad, "/", 48)
# also top-p quick check
P=0.7
tp2 = torch.full((B,),P,device="cuda"); tk2=torch.zeros(B,dtype=torch.int32,device="cuda")
logits2=(torch.randn(B,V,generator=g)*2).cuda()
probs=torch.softmax(logits2.float(),dim=-1); sp,si=probs.sort(dim=-1,descending=True); cum=sp.cumsum(-1)
badp=0
for s in range(6):
    steps = torch.full((B,),s,dtype=torch.int64,device="cuda")
    toks,_=ops.sample(logits2,torch.ones(B,device="cuda"),tp2,tk2,seeds,steps)
    for b in range(B):
        ncut=int((cum[b]<P).sum().item())+1
        if int(toks[b]) not in set(si[b,:ncut].tolist()): badp+=1
print("bad nucleus draws:", badp)

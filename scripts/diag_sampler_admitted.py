"""Diagnostic: empirically probe the fused sampler's ADMITTED set.

Draws many steps at fixed seeds and collects the distinct tokens drawn per
row — with the Gumbel-max draw, the drawn-set over hundreds of steps equals
the admitted set. Used to verify exact top-k on hardware (found the
fp-contract sx inconsistency fixed in sampling.hip); the kernel's optional
16-float debug buffer prints the refinement walk per row.
"""
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
drawn = [set() for _ in range(B)]
for s in range(800):
    steps = torch.full((B,),s,dtype=torch.int64,device="cuda")
    toks,_ = ops.sample(logits,t,tp,tk,seeds,steps)
    for b in range(B):
        drawn[b].add(int(toks[b]))
for b in range(B):
    sv, si = base[b].sort(descending=True)
    ranks = sorted(int((si==tok).nonzero()[0]) for tok in drawn[b])
    print(f"b={b}: admitted>= {len(drawn[b])} ranks={ranks}")

# debug pass: one call with diagnostics buffer
dbg = torch.zeros(B, 16, device="cuda")
steps = torch.zeros(B, dtype=torch.int64, device="cuda")
toks,_ = ops.sample(logits,t,tp,tk,seeds,steps,None,dbg)
d = dbg.cpu()
for b in range(B):
    print(f"b={b} coarse(cut,cnt,need,state)={d[b,:4].tolist()} "
          f"it0={d[b,4:8].tolist()} it1={d[b,8:12].tolist()} it2={d[b,12:16].tolist()}")

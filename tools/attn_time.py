import torch, sys
sys.path.insert(0, ".")
from code2vec_amd.ops import ext
dev = "cuda:0"
B, C, EP, E = 1024, 200, 128, 100
g = torch.Generator().manual_seed(3)
ccv = (torch.randn(B, C, EP, generator=g)*0.5).to(dev, torch.bfloat16)
a = torch.randn(EP, generator=g).to(dev); a[E:] = 0
starts = torch.randint(1, 100, (B, C), generator=g, dtype=torch.int32).to(dev)
cv = torch.empty(B, EP, device=dev); attn = torch.empty(B, C, device=dev)
dccv = torch.empty(B, C, EP, dtype=torch.bfloat16, device=dev)
da = torch.empty(B, EP, device=dev)
dcv = torch.randn(B, EP, generator=g).to(dev)
none = torch.Tensor()
def t(f, n=200):
    for _ in range(20): f()
    torch.cuda.synchronize(); s,e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(n): f()
    e.record(); torch.cuda.synchronize(); return s.elapsed_time(e)/n*1000
print("fwd us:", round(t(lambda: ext().attention_fwd(ccv, a, starts, cv, attn, E)), 1))
print("bwd us:", round(t(lambda: ext().attention_bwd(dcv, none, ccv, a, starts, attn, dccv, da, E, False)), 1))

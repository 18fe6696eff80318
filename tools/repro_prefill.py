import sys, time
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd import ops
ext = ops.hip_ext()

def ck(name):
    torch.cuda.synchronize(); print("ok:", name, flush=True)

torch.manual_seed(0)
E, H, I, K = 64, 2048, 1408, 6
N = 16384
dev = "cuda"
x = torch.randn(N, H, dtype=torch.bfloat16, device=dev)
gate_w = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev) * 0.03
up_w = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev) * 0.03
down_w = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) * 0.03
weights = torch.rand(N, K, dtype=torch.bfloat16, device=dev)
indices = torch.randint(0, E, (N, K), device=dev)
ck("init")

flat_e = indices.reshape(-1).long()
order = torch.argsort(flat_e, stable=True)
sorted_e = flat_e[order]
tok = torch.arange(N, device=dev).repeat_interleave(K)[order]
wts = weights.reshape(-1).float()[order]
counts = torch.zeros(E, dtype=torch.long, device=dev)
counts.scatter_add_(0, sorted_e, torch.ones_like(sorted_e))
start = torch.cumsum(counts, 0) - counts
cap = (int(counts.max()) + 127) // 128 * 128
P = N * K
print("cap", cap, "P", P)
slot = torch.arange(P, device=dev) - start[sorted_e]
dst = sorted_e * cap + slot
print("dst range", int(dst.min()), int(dst.max()), "limit", E * cap)
ck("indexing")
xp = x.new_zeros(E * cap, H)
xp[dst] = x[tok]
xp = xp.view(E, cap, H)
ck("scatter")
g = torch.bmm(xp, gate_w.transpose(1, 2))
ck("bmm g")
u = torch.bmm(xp, up_w.transpose(1, 2))
ck("bmm u")
hh = ops.swiglu(g, u)
ck("swiglu")
d = torch.bmm(hh, down_w.transpose(1, 2)).reshape(E * cap, H)
ck("bmm d")
y = d[dst].float() * wts[:, None]
out = torch.zeros(N, H, device=dev, dtype=torch.float32)
out.index_add_(0, tok, y)
ck("gather")
print("done", torch.isfinite(out).all().item())

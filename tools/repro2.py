import sys, time
sys.path.insert(0, "/root/repo")
import torch
def ck(n):
    torch.cuda.synchronize(); print("ok:", n, flush=True)
E, cap, H, I = 64, 1664, 2048, 1408
xp = torch.randn(E, cap, H, dtype=torch.bfloat16, device="cuda")
w = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda")
ck("init")
wt = w.transpose(1, 2).contiguous()
ck("contig transpose")
g = torch.bmm(xp, wt)
ck("bmm contiguous-B")
g2 = torch.matmul(xp, w.transpose(1, 2))
ck("matmul transposed-B")
g3 = torch.bmm(xp, w.transpose(1, 2))
ck("bmm transposed-B")
print("all good", g.shape)

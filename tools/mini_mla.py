import sys
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd import ops
ext = ops.hip_ext()
B, T, nh, nope, vd, rope, Scap = 2, 3, 4, 32, 32, 16, 64
kvh = torch.randn(B, T, nh, nope + vd, dtype=torch.bfloat16, device="cuda")
kpe = torch.randn(B, T, rope, dtype=torch.bfloat16, device="cuda")
kc = torch.zeros(B, nh, Scap, nope + rope, dtype=torch.bfloat16, device="cuda")
vc = torch.zeros(B, nh, Scap, vd, dtype=torch.bfloat16, device="cuda")
ext.mla_append_kv(kvh, kpe, kc, vc, pos0=5)
torch.cuda.synchronize()
print("basic ok")
# verify
for b in range(B):
    for t in range(T):
        for h in range(nh):
            assert torch.equal(kc[b, h, 5 + t, :nope], kvh[b, t, h, :nope])
            assert torch.equal(kc[b, h, 5 + t, nope:], kpe[b, t])
            assert torch.equal(vc[b, h, 5 + t], kvh[b, t, h, nope:])
print("values ok")
pos = torch.tensor([7], dtype=torch.int32, device="cuda")
ext.mla_append_kv(kvh[:, :1], kpe[:, :1], kc, vc, pos=pos)
torch.cuda.synchronize()
assert torch.equal(kc[0, 0, 7, :nope], kvh[0, 0, 0, :nope])
print("graph-pos ok")

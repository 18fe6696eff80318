# Forensic script for the attn_prefill silent-skip bug (kept as a record:
# found Dk=Dv=32 fell through the launcher and returned uninitialized output).
import sys, pathlib, json
sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch
from safetensors.torch import save_file
from mlx_sharding_amd.models import get_model_class
from mlx_sharding_amd.utils.presets import get_preset
from mlx_sharding_amd.utils.loading import load_model
from mlx_sharding_amd.parallel import wire

cfg = get_preset("debug-llama")
cls = get_model_class("llama")
torch.manual_seed(13)
m = cls(cfg, cfg.shard(0, cfg.num_hidden_layers))
for p in m.parameters():
    p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
d = pathlib.Path("/tmp/dbg_ckpt"); d.mkdir(exist_ok=True)
save_file({k: v for k, v in m.state_dict().items() if "rope_inv_freq" not in k},
          str(d / "model.safetensors"))
json.dump(cfg.raw, open(d / "config.json", "w"))

ids = torch.randint(0, cfg.vocab_size, (1, 6),
                    generator=torch.Generator().manual_seed(3)).cuda()
mf, _ = load_model(d, device="cuda")
m0, _ = load_model(d, 0, 2, device="cuda")
m1, _ = load_model(d, 2, 4, device="cuda")
with torch.no_grad():
    full = mf(ids, mf.make_cache())
    h = m0(ids, m0.make_cache())
    out = m1(h, m1.make_cache())
    # wire roundtrip variant
    h2 = wire.msg_to_tensor(wire.tensor_to_msg(h), device="cuda")
    out2 = m1(h2.to(h.dtype), m1.make_cache())
torch.cuda.synchronize()
a, b, c = full[:, -1].float(), out[:, -1].float(), out2[:, -1].float()
print("full vs chain maxdiff:", (a - b).abs().max().item(),
      "argmax:", a.argmax(-1).item(), b.argmax(-1).item())
print("chain vs wire-chain maxdiff:", (b - c).abs().max().item())
# locate divergence: compare h against full model's intermediate
cache = mf.make_cache()
x = mf.model.embed_tokens(ids)
cos, sin, _ = mf.rope_for(cache[0], x.shape[1], x.device)
for j, i in enumerate(range(4)):
    x = mf.model.layers[str(i)](x, cos, sin, cache[j])
    if i == 1:
        print("h after layer1: maxdiff vs stage0 out:",
              (x.float() - h.float()).abs().max().item())

# --- narrow down: run mf's layer objects vs m1's on the SAME h -------------
with torch.no_grad():
    cA = mf.make_cache()
    cB = m1.make_cache()
    cosA, sinA, _ = mf.rope_for(cA[0], 6, h.device)
    cosB, sinB, _ = m1.rope_for(cB[0], 6, h.device)
    print("cos tables equal:", torch.equal(cosA, cosB),
          torch.equal(sinA, sinB))
    xA = h.clone(); xB = h.clone()
    for i in (2, 3):
        xA = mf.model.layers[str(i)](xA, cosA, sinA, cA[i])
        xB = m1.model.layers[str(i)](xB, cosB, sinB, cB[i - 2])
        print(f"layer {i} out maxdiff:", (xA.float() - xB.float()).abs().max().item())
    nA = mf.model.norm(xA); nB = m1.model.norm(xB)
    print("norm maxdiff:", (nA.float() - nB.float()).abs().max().item())
    print("norm w equal:", torch.equal(mf.model.norm.weight, m1.model.norm.weight))
    print("lm_head w equal:", torch.equal(mf.lm_head.weight, m1.lm_head.weight))
    lA = mf.lm_head(nA); lB = m1.lm_head(nB)
    print("lm maxdiff:", (lA.float() - lB.float()).abs().max().item())
    # weights of layer 2 equal?
    s1 = mf.model.layers["2"].state_dict(); s2 = m1.model.layers["2"].state_dict()
    for k in s1:
        if not torch.equal(s1[k], s2[k]):
            print("WEIGHT DIFF:", k, (s1[k].float()-s2[k].float()).abs().max().item())

# --- determinism check: same model, same input, twice ----------------------
with torch.no_grad():
    r1 = mf(ids, mf.make_cache())
    r2 = mf(ids, mf.make_cache())
    print("full twice maxdiff:", (r1.float() - r2.float()).abs().max().item())
    l2 = mf.model.layers["2"]
    c1 = mf.make_cache(); c2 = mf.make_cache()
    y1 = l2(h, cosA, sinA, c1[2]); y2 = l2(h, cosA, sinA, c2[2])
    print("layer2 twice maxdiff:", (y1.float() - y2.float()).abs().max().item())
    w = l2.self_attn._fused_qkv
    g1 = torch.nn.functional.linear(h, w.weight)
    g2 = torch.nn.functional.linear(h, w.weight)
    print("fused qkv GEMM twice maxdiff:", (g1.float() - g2.float()).abs().max().item())

# --- find the eerste diverging layer across two full runs ------------------
acts = {}
def mk_hook(tag, store):
    def hook(mod, inp, out):
        store.append(out.detach().clone() if torch.is_tensor(out) else None)
    return hook

for run in (0, 1):
    store = []
    hs = [mf.model.layers[str(i)].register_forward_hook(mk_hook(i, store))
          for i in range(4)]
    eh = mf.model.embed_tokens.register_forward_hook(mk_hook("e", store))
    with torch.no_grad():
        mf(ids, mf.make_cache())
    for hdl in hs: hdl.remove()
    eh.remove()
    acts[run] = store
for i, (a, b) in enumerate(zip(acts[0], acts[1])):
    if a is not None:
        print(f"mod {i}: twice maxdiff {(a.float()-b.float()).abs().max().item()}")

# --- uninit-read hunt: zero the cache backing buffers ----------------------
from mlx_sharding_amd.ops import kvcache as KC
orig_ensure = KC.KVCache._ensure
def zero_ensure(self, batch, needed):
    orig_ensure(self, batch, needed)
    # freshly ensured buffers: zero everything past offset
    if self._k is not None:
        self._k[:, :, self.offset:].zero_()
        self._v[:, :, self.offset:].zero_()
KC.KVCache._ensure = zero_ensure
with torch.no_grad():
    z1 = mf(ids, mf.make_cache())
    z2 = mf(ids, mf.make_cache())
print("zeroed-cache full twice maxdiff:",
      (z1.float() - z2.float()).abs().max().item())
KC.KVCache._ensure = orig_ensure
# also: does r1-style nondeterminism persist without zeroing, interleaved
# with an allocation-pattern scrambler?
with torch.no_grad():
    a1 = mf(ids, mf.make_cache())
    junk = [torch.randn(37 + i, 113, device="cuda") for i in range(40)]
    del junk
    a2 = mf(ids, mf.make_cache())
print("scrambled-alloc full twice maxdiff:",
      (a1.float() - a2.float()).abs().max().item())

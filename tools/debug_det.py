# Forensic script for the allocator-address-dependent nondeterminism hunt
# (led to the attn_prefill silent-skip fix).
import sys, pathlib, json
sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch
from safetensors.torch import save_file
from mlx_sharding_amd.models import get_model_class
from mlx_sharding_amd.utils.presets import get_preset
from mlx_sharding_amd.utils.loading import load_model

cfg = get_preset("debug-llama")
cls = get_model_class("llama")
torch.manual_seed(13)
m = cls(cfg, cfg.shard(0, cfg.num_hidden_layers))
for p in m.parameters():
    p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
d = pathlib.Path("/tmp/dbg_ckpt2"); d.mkdir(exist_ok=True)
save_file({k: v for k, v in m.state_dict().items() if "rope_inv_freq" not in k},
          str(d / "model.safetensors"))
json.dump(cfg.raw, open(d / "config.json", "w"))
ids = torch.randint(0, cfg.vocab_size, (1, 6),
                    generator=torch.Generator().manual_seed(3)).cuda()
mf, _ = load_model(d, device="cuda")
import os
print("prefer_hipblaslt env:", os.environ.get("TORCH_BLAS_PREFER_HIPBLASLT"))
outs = []
for i in range(6):
    with torch.no_grad():
        outs.append(mf(ids, mf.make_cache()).float().clone())
    if i == 2:  # scramble the allocator mid-sequence
        junk = [torch.randn(31 + 7 * j, 97, device="cuda") for j in range(60)]
        del junk
m0 = outs[0]
for i, o in enumerate(outs[1:], 1):
    print(f"run {i} vs 0 maxdiff: {(o - m0).abs().max().item()}")

# --- locate: per-module capture with a scramble between runs --------------
acts = {}
for run in (0, 1):
    store = []
    def mk(store):
        def hook(mod, inp, out):
            store.append(out.detach().float().clone()
                         if torch.is_tensor(out) else None)
        return hook
    hs = [mf.model.layers[str(i)].register_forward_hook(mk(store))
          for i in range(4)]
    hs.append(mf.model.embed_tokens.register_forward_hook(mk(store)))
    hs.append(mf.model.norm.register_forward_hook(mk(store)))
    with torch.no_grad():
        mf(ids, mf.make_cache())
    for h in hs:
        h.remove()
    acts[run] = store
    junk = [torch.randn(13 + 11 * j, 53, device="cuda") for j in range(77)]
    del junk
for i, (a, b) in enumerate(zip(acts[0], acts[1])):
    if a is not None:
        print(f"mod#{i}: scramble maxdiff {(a - b).abs().max().item()}")

# finer: inside layer 0, capture attention sub-steps
l0 = mf.model.layers["0"]
subs = {}
for run in (0, 1):
    store = []
    hs = [l0.self_attn.register_forward_hook(mk(store)),
          l0.mlp.register_forward_hook(mk(store)),
          l0.input_layernorm.register_forward_hook(mk(store))]
    with torch.no_grad():
        mf(ids, mf.make_cache())
    for h in hs:
        h.remove()
    subs[run] = store
    junk = [torch.randn(17 + 13 * j, 41, device="cuda") for j in range(66)]
    del junk
names = ["input_ln", "self_attn", "mlp"] * 4
for i, (a, b) in enumerate(zip(subs[0], subs[1])):
    if a is not None and i < 6:
        print(f"l0-sub {names[i % 3]} #{i}: maxdiff {(a - b).abs().max().item()}")

# --- op-level tracer across a diverging pair ------------------------------
from mlx_sharding_amd import ops as OPS
trace = {}
def wrap(name, fn):
    def inner(*a, **k):
        out = fn(*a, **k)
        tl = trace.setdefault(name, [])
        if torch.is_tensor(out):
            tl.append(out.detach().float().clone())
        elif isinstance(out, tuple):
            tl.append(tuple(o.detach().float().clone() for o in out))
        return out
    return inner

orig = {n: getattr(OPS, n) for n in
        ("rms_norm", "rms_norm_residual", "apply_rope", "attention",
         "swiglu", "linear")}
runs = []
for run in (0, 1):
    trace.clear()
    for n, f in orig.items():
        setattr(OPS, n, wrap(n, f))
    with torch.no_grad():
        mf(ids, mf.make_cache())
    for n, f in orig.items():
        setattr(OPS, n, f)
    runs.append({n: list(v) for n, v in trace.items()})
    junk = [torch.randn(19 + 23 * j, 37, device="cuda") for j in range(55)]
    del junk
for n in orig:
    a, b = runs[0].get(n, []), runs[1].get(n, [])
    for i, (x1, x2) in enumerate(zip(a, b)):
        t1 = x1 if torch.is_tensor(x1) else x1[0]
        t2 = x2 if torch.is_tensor(x2) else x2[0]
        dmax = (t1 - t2).abs().max().item()
        if dmax > 0:
            print(f"FIRST-DIVERGING op {n} call#{i}: {dmax}")
            break
    else:
        print(f"{n}: all {len(a)} calls identical")

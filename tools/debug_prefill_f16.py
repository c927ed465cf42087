#!/usr/bin/env python3
"""Isolate the f16 _mt prefill divergence: run the same T-token prefill
through (a) the 64-token tile path (known good) and (b) the _mt path,
compare the residual stream AND the KV caches layer by layer."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from distributedllm_amd.engine import HIPSliceEngine, TorchSliceEngine
from distributedllm_amd.formats import ggml, slicer, synthetic

ftype = int(sys.argv[1]) if len(sys.argv) > 1 else ggml.FTYPE_MOSTLY_F16
T = int(sys.argv[2]) if len(sys.argv) > 2 else 130

f = synthetic.build_model("small", ftype=ftype, seed=0)
ex = slicer.make_extra_layers(f)

eng_mt = HIPSliceEngine.from_ggml(f, n_ctx=256, max_batch=1)
eng_mt.attach_extra(ex)
eng_tile = HIPSliceEngine.from_ggml(f, n_ctx=256, max_batch=1)
eng_tile.attach_extra(ex)
# force the tile path on the reference engine
eng_tile._mfma_path = lambda: False

hp = f.hparams
torch.manual_seed(13)
x = (torch.randn(T, hp.n_embd) * 0.5).cuda()
pos = torch.arange(T, dtype=torch.int32).cuda()
seq = torch.zeros(T, dtype=torch.int32).cuda()

y_mt = eng_mt.forward(x.clone(), pos, seq)
y_tile = eng_tile.forward(x.clone(), pos, seq)
torch.cuda.synchronize()


def rel(a, b):
    return ((a - b).norm() / b.norm().clamp_min(1e-9)).item()


print("y rel:", rel(y_mt.float(), y_tile.float()))
kc_mt, vc_mt = eng_mt._eng.k_cache, eng_mt._eng.v_cache
kc_t, vc_t = eng_tile._eng.k_cache, eng_tile._eng.v_cache
L = kc_mt.shape[0]
for li in range(L):
    km = kc_mt[li, 0, :T].float()
    kt = kc_t[li, 0, :T].float()
    vm = vc_mt[li, 0, :T].float()
    vt = vc_t[li, 0, :T].float()
    print(f"layer {li}: k rel {rel(km, kt):.3e}  v rel {rel(vm, vt):.3e}")
    if rel(km, kt) > 1e-3:
        d = (km - kt).abs().amax(dim=1)
        bad = (d > d.max() * 0.5).nonzero()[:, 0].tolist()
        print("   bad k rows (pos):", bad[:20], "of", len(bad))
        e = (km - kt).abs().amax(dim=0)
        bade = (e > e.max() * 0.5).nonzero()[:, 0].tolist()
        print("   bad k cols (E):", bade[:20], "of", len(bade))
        break
# token-wise divergence of the residual stream
d = (y_mt.float() - y_tile.float()).abs().amax(dim=1)
bad = (d > max(d.max().item(), 1e-9) * 0.5).nonzero()[:, 0].tolist()
print("bad y rows:", bad[:30], "of", len(bad))

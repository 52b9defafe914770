"""Bisect the hipGraph-capture divergence: forward-only, fwd+bwd, full step."""
import os
import torch

from marian_amd.common.options import Options
from marian_amd.data.synthetic import synthetic_batch
from marian_amd.models.model_factory import create_model
from marian_amd.training.graph_group import SyncGraphGroup

DEV = "cuda"
O = {
    "type": "transformer", "dim-emb": 128, "enc-depth": 2, "dec-depth": 2,
    "transformer-heads": 4, "transformer-dim-ffn": 256,
    "transformer-preprocess": "", "transformer-postprocess": "dan",
    "transformer-postprocess-emb": "d", "transformer-postprocess-top": "",
    "transformer-ffn-activation": "relu", "transformer-dropout": 0.0,
    "dim-vocabs": [1000, 1000], "tied-embeddings-all": True,
    "transformer-train-position-embeddings": False,
    "transformer-decoder-dim-ffn": 0,
    "precision": ["bfloat16", "float32"],
    "cost-type": "ce-mean-words", "label-smoothing": 0.1,
    "optimizer": "adam", "learn-rate": 1e-2, "clip-norm": 0.0,
}


def fresh():
    torch.manual_seed(7)
    opts = Options(dict(O))
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    gg.set_lr(1e-2)
    return gg


def params(gg):
    return torch.cat([b.flat_param[:b.used].float()
                      for b in gg.store.buckets]).clone()


b0 = synthetic_batch(16, 12, 12, 1000, seed=100, device=DEV)
b1 = synthetic_batch(16, 12, 12, 1000, seed=101, device=DEV)

# ---- 1. forward-only capture --------------------------------------------
gg = fresh()
with torch.no_grad():
    ref0 = gg.model(b0).float().clone()
    ref1 = gg.model(b1).float().clone()
static = b0.to(torch.device(DEV))
s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    with torch.no_grad():
        for _ in range(3):
            gg.model(static)
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    with torch.no_grad():
        out = gg.model(static)
g.replay()
torch.cuda.synchronize()
print("fwd replay b0 err:", (out.float() - ref0).abs().max().item())
static.source.indices.copy_(b1.source.indices)
static.source.mask.copy_(b1.source.mask)
static.target.indices.copy_(b1.target.indices)
static.target.mask.copy_(b1.target.mask)
g.replay()
torch.cuda.synchronize()
print("fwd replay b1 err:", (out.float() - ref1).abs().max().item())

# ---- 2. fwd+bwd capture: compare gradients ------------------------------
gg_e = fresh()
gg_g = fresh()


def fwd_bwd(gg, batch):
    logits = gg.model(batch)
    lv = gg.criterion(logits, batch)
    lv.ce_sum.backward()
    return lv


# eager grads on b1 after 1 prior fwd/bwd on b0 (buffers warm)
for bt in (b0,):
    fwd_bwd(gg_e, bt)
    gg_e.store.zero_grad()
fwd_bwd(gg_e, b1)
grads_e = torch.cat([b.flat_grad[:b.used].float() for b in gg_e.store.buckets]).clone()

static2 = b0.to(torch.device(DEV))
s2 = torch.cuda.Stream()
s2.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s2):
    for _ in range(3):
        fwd_bwd(gg_g, static2)
        gg_g.store.zero_grad()
torch.cuda.current_stream().wait_stream(s2)
torch.cuda.synchronize()
g2 = torch.cuda.CUDAGraph()
with torch.cuda.graph(g2):
    fwd_bwd(gg_g, static2)
# replay on b1
static2.source.indices.copy_(b1.source.indices)
static2.source.mask.copy_(b1.source.mask)
static2.target.indices.copy_(b1.target.indices)
static2.target.mask.copy_(b1.target.mask)
gg_g.store.zero_grad()
torch.cuda.synchronize()
g2.replay()
torch.cuda.synchronize()
grads_g = torch.cat([b.flat_grad[:b.used].float() for b in gg_g.store.buckets]).clone()
d = (grads_e - grads_g).abs()
print("bwd replay grad err:", d.max().item(), "rel:",
      (d.max() / grads_e.abs().max()).item())

# ---- 3. full step capture ------------------------------------------------
gg3 = fresh()
p_before = params(gg3)
fn = gg3.make_graphed_update(b0, warmup=2)
gg4 = fresh()
for _ in range(2):
    gg4.update(b0)
lv4 = gg4.update(b1)
lv3 = fn(b1)
torch.cuda.synchronize()
print("step ce eager", float(lv4.ce_sum), "graph", float(lv3.ce_sum))
print("step param err:", (params(gg3) - params(gg4)).abs().max().item())

# ---- 4. replicate the failing test exactly, with probes ------------------
print("=== test replication ===")
batches = [synthetic_batch(16, 12, 12, 1000, seed=200 + i, device=DEV)
           for i in range(4)]

def run(graphed):
    torch.manual_seed(7)
    opts = Options(dict(O))
    model = create_model(opts)
    gg = SyncGraphGroup(opts, model, torch.device(DEV))
    gg.set_lr(1e-2)
    if graphed:
        fn = gg.make_graphed_update(batches[0], warmup=2)
    else:
        for _ in range(2):
            gg.update(batches[0])
        fn = gg.update
    torch.cuda.synchronize()
    print("  post-warmup param sum:", params(gg).double().sum().item())
    out = []
    for i, b in enumerate(batches):
        gg.set_lr(1e-2 * (1.0 + 0.5 * i))
        lv = fn(b)
        torch.cuda.synchronize()
        out.append(float(lv.ce_sum))
        print(f"  step{i} ce={float(lv.ce_sum):.2f} psum={params(gg).double().sum().item():.4f}")
    return out, params(gg)

e_out, e_p = run(False)
g_out, g_p = run(True)
print("param maxdiff:", (e_p - g_p).abs().max().item())

import sys, time, torch
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import ops

def sync(): torch.cuda.synchronize()

cfg = EngineConfig.preset("bert", compressor="dense", dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
flat = torch.randn(109_500_000, device="cuda")
res = torch.zeros_like(flat)
n = flat.numel()

def s1():  # ef + bump-count + compact of t
    ops.ef_restore_snapshot_(flat, res)
    ops.count_multi_gt(flat, [0.3,0.31,0.32])
    idx, val = ops.compact_gt(flat, 0.3)
    return idx, val

def s2():
    idx, val = s1()
    reduced = torch.zeros(n, device="cuda")
    ops.scatter_add_(reduced, idx, val)
    return reduced

def s3():
    reduced = s2()
    gi, gv = ops.compact_gt(reduced, 0.5)
    pack = torch.cat([gi.view(torch.int32), gv.view(torch.int32)])
    buf = pack.clone()
    return gi, gv

def s4():
    gi, gv = s3()
    out = flat  # fill in place like the engine does
    ops.fill_sparse_scaled_(out, gi, gv, 1.0)
    mask = torch.zeros(n, dtype=torch.bool, device="cuda")
    mask[gi.long()] = True
    member = mask[gi.long()]
    inv = gi[member]
    ops.zero_at_(res, inv)

which = sys.argv[1]
fn = {"s1": s1, "s2": s2, "s3": s3, "s4": s4, "none": lambda: None}[which]
for _ in range(3): tr.step(); fn()
out=[]
for i in range(12):
    sync(); t0=time.perf_counter()
    tr.step(); fn()
    sync(); out.append(1000*(time.perf_counter()-t0))
print(which, " ".join(f"{x:.1f}" for x in out))

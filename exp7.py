import time, torch
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import ops, allreducer

def sync(): torch.cuda.synchronize()

# monkeypatch _oktopk to log sizes
orig = allreducer.AllReducer._oktopk
log = []
def patched(self, name, t, st):
    idx_sizes = {}
    orig_compact = ops.compact_gt
    def wrap(tt, tau):
        i, v = orig_compact(tt, tau)
        idx_sizes.setdefault('sizes', []).append(i.numel())
        return i, v
    ops.compact_gt = wrap
    try:
        out = orig(self, name, t, st)
    finally:
        ops.compact_gt = orig_compact
    log.append((st.tau_local, st.tau_global, idx_sizes.get('sizes', [])))
    return out
allreducer.AllReducer._oktopk = patched

cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
for _ in range(3): tr.step()
for i in range(12):
    sync(); t0=time.perf_counter()
    tr.opt.zero_grad()
    loss = tr._forward_loss()
    loss.backward()
    sync(); t1=time.perf_counter()
    tr.opt.step()
    sync(); t2=time.perf_counter()
    tl, tg, sizes = log[-1]
    print(f"step{i}: fwdbwd {1000*(t1-t0):6.1f} opt {1000*(t2-t1):5.1f} tau_l {tl:.3e} tau_g {tg:.3e} compacts {sizes}")

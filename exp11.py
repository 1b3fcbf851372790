import time, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd import AllReducer, Comm, ops
def sync(): torch.cuda.synchronize()

# wrap every op with synced timing
import oktopk_amd.allreducer as AR
times = {}
def wrap(name, fn):
    def g(*a, **k):
        sync(); t0=time.perf_counter()
        r = fn(*a, **k)
        sync(); times[name] = times.get(name,0)+1000*(time.perf_counter()-t0)
        return r
    return g
for fname in ["compact_gt","count_multi_gt","kth_abs_value","scatter_add_","zero_at_",
              "fill_sparse_scaled_","ef_restore_snapshot_","isin_sorted","l2norm"]:
    setattr(AR.ops, fname, wrap(fname, getattr(ops, fname)))

eng = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                 oktopk=OkTopkConfig(dense_warmup_iters=0)))
flat = torch.randn(109_500_000, device="cuda")
for i in range(40):
    times.clear()
    sync(); t0=time.perf_counter()
    eng.run("x", flat)
    sync(); dt=1000*(time.perf_counter()-t0)
    acc = sum(times.values())
    if dt > 10 or i < 3:
        top = sorted(times.items(), key=lambda x:-x[1])[:4]
        print(f"call {i}: {dt:6.1f} ms (ops {acc:5.1f})", " ".join(f"{k}={v:.1f}" for k,v in top))

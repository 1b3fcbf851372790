import time, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import AllReducer, Comm
def sync(): torch.cuda.synchronize()

# A) engine alone, 40 calls
eng = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                 oktopk=OkTopkConfig(dense_warmup_iters=0)))
flat = torch.randn(109_500_000, device="cuda")
times=[]
for i in range(40):
    sync(); t0=time.perf_counter()
    eng.run("x", flat)
    sync(); times.append(1000*(time.perf_counter()-t0))
print("engine alone:", " ".join(f"{x:.1f}" for x in times))

# B) dense trainer + engine on side tensor every 2nd step
cfg = EngineConfig.preset("bert", compressor="dense", dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
eng2 = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                 oktopk=OkTopkConfig(dense_warmup_iters=0)))
flat2 = torch.randn(109_500_000, device="cuda")
for _ in range(3): tr.step()
out=[]
for i in range(18):
    sync(); t0=time.perf_counter()
    tr.step()
    if i % 2 == 0:
        eng2.run("y", flat2)
    sync(); out.append(1000*(time.perf_counter()-t0))
print("every-2nd:", " ".join(f"{x:.1f}" for x in out))

import time, torch
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer
def sync(): torch.cuda.synchronize()
cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
for _ in range(3): tr.step()
out=[]
for i in range(12):
    sync(); t0=time.perf_counter()
    tr.step()
    sync(); out.append(1000*(time.perf_counter()-t0))
print(" ".join(f"{x:.1f}" for x in out))

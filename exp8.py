import time, torch
from torch.profiler import profile, ProfilerActivity
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer

def sync(): torch.cuda.synchronize()

cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
for _ in range(4): tr.step()

for i in range(6):
    sync(); t0=time.perf_counter()
    with profile(activities=[ProfilerActivity.CPU]) as prof:
        tr.opt.zero_grad()
        loss = tr._forward_loss()
        loss.backward()
    sync(); dt=1000*(time.perf_counter()-t0)
    tr.opt.step()
    if dt > 30:
        print(f"SPIKY step {i}: {dt:.1f} ms")
        print(prof.key_averages().table(sort_by="self_cpu_time_total", row_limit=12))
        break
    else:
        print(f"step {i}: {dt:.1f} ms")

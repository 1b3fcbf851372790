import time, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd.trainer import Trainer

def sync(): torch.cuda.synchronize()

cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
for _ in range(3): tr.step()
prev = torch.cuda.memory_stats()
for i in range(10):
    sync(); t0=time.perf_counter()
    tr.step()
    sync(); dt=1000*(time.perf_counter()-t0)
    st = torch.cuda.memory_stats()
    da = st["num_device_alloc"]-prev["num_device_alloc"]
    df = st["num_device_free"]-prev["num_device_free"]
    seg = st["segment.all.current"]
    res = st["reserved_bytes.all.current"]/2**30
    prev = st
    print(f"step {i}: {dt:6.1f} ms  hipMalloc {da:3d} hipFree {df:3d} segments {seg} reserved {res:.2f} GiB")

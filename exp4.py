import time, torch
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import ops

def sync(): torch.cuda.synchronize()

cfg = EngineConfig.preset("bert", compressor="dense", dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
flat = torch.randn(109_500_000, device="cuda")
res = torch.zeros_like(flat)

variants = {
    "nothing": lambda: None,
    "count_multi": lambda: ops.count_multi_gt(flat, [0.1,0.11,0.12,0.13,0.14,0.15]),
    "compact": lambda: ops.compact_gt(flat, 0.2),
    "ef_restore": lambda: ops.ef_restore_snapshot_(flat, res),
    "kth": lambda: ops.kth_abs_value(flat, 100000),
    "zeros110M": lambda: torch.zeros_like(flat).sum().item(),
    "small_d2h": lambda: [flat[:4].cpu() for _ in range(6)],
}
for _ in range(3): tr.step()
for name, fn in variants.items():
    out=[]
    for i in range(9):
        sync(); t0=time.perf_counter()
        tr.step(); fn()
        sync(); out.append(1000*(time.perf_counter()-t0))
    print(f"{name:12s}", " ".join(f"{x:.1f}" for x in out))

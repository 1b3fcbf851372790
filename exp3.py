import gc, time, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd.trainer import Trainer

def sync(): torch.cuda.synchronize()

def run(label, disable_gc):
    cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
    tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
    for _ in range(3): tr.step()
    if disable_gc: gc.disable()
    counts0 = gc.get_stats()
    out=[]
    for i in range(12):
        sync(); t0=time.perf_counter()
        tr.step()
        sync(); out.append(1000*(time.perf_counter()-t0))
    if disable_gc: gc.enable()
    print(label, " ".join(f"{x:.1f}" for x in out))
    print("  gc stats delta:", [ (a['collections']-b['collections']) for a,b in zip(gc.get_stats(), counts0)])

run("gc-on ", False)
run("gc-off", True)

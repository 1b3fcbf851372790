"""Experiment: where do the sparse-path milliseconds go? (scratch, not product)"""
import time, sys, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import AllReducer, Comm

def sync():
    torch.cuda.synchronize()

def phase_timed_steps(tr, n=6, label=""):
    per=[]
    for _ in range(n):
        tr.opt.zero_grad()
        sync(); t0 = time.perf_counter()
        loss = tr._forward_loss()
        sync(); t1 = time.perf_counter()
        loss.backward()
        sync(); t2 = time.perf_counter()
        tr.opt.step()
        sync(); t3 = time.perf_counter()
        per.append((1000*(t1-t0),1000*(t2-t1),1000*(t3-t2)))
    fw=sum(p[0] for p in per)/n; bw=sum(p[1] for p in per)/n; op=sum(p[2] for p in per)/n
    print(f"{label}: fwd {fw:.2f} bwd {bw:.2f} opt {op:.2f} ms")
    print("   per-step:", " ".join(f"({a:.1f},{b:.1f},{c:.1f})" for a,b,c in per))

def main():
    dev = torch.device("cuda")
    # B: engine.run alone on a 110M flat tensor
    eng = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                     oktopk=OkTopkConfig(dense_warmup_iters=0)))
    flat = torch.randn(109_500_000, device=dev)
    for i in range(3):
        eng.run("flat", flat)
    sync(); t0 = time.perf_counter()
    for i in range(5):
        eng.run("flat", flat)
    sync()
    print(f"engine.run alone: {1000*(time.perf_counter()-t0)/5:.2f} ms/call")

    for comp in ("oktopk", "dense"):
        cfg = EngineConfig.preset("bert", compressor=comp, density=0.001,
                                  dense_warmup_iters=0)
        tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
        for _ in range(3):
            tr.step()
        phase_timed_steps(tr, 6, comp)
        del tr
        torch.cuda.empty_cache()

    # C: dropout-free
    cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001,
                              dense_warmup_iters=0)
    tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16",
                 model_kwargs=dict(hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0))
    for _ in range(3):
        tr.step()
    phase_timed_steps(tr, 6, "oktopk-nodropout")

main()

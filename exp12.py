import threading, time, sys, traceback, collections, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd import AllReducer, Comm
def sync(): torch.cuda.synchronize()

samples = collections.Counter()
running = True
main_tid = threading.get_ident()
def sampler():
    while running:
        frames = sys._current_frames()
        f = frames.get(main_tid)
        if f is not None:
            stack = traceback.extract_stack(f)
            # innermost 3 frames
            key = " <- ".join(f"{fr.filename.split('/')[-1]}:{fr.lineno}:{fr.name}" for fr in stack[-3:])
            samples[key] += 1
        time.sleep(0.004)
th = threading.Thread(target=sampler, daemon=True); th.start()

eng = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                 oktopk=OkTopkConfig(dense_warmup_iters=0)))
flat = torch.randn(109_500_000, device="cuda")
for i in range(5): eng.run("x", flat)
sync()
samples.clear()
for i in range(40):
    eng.run("x", flat)
sync()
running = False
for k, v in samples.most_common(12):
    print(f"{v:5d}  {k}")

import time, torch
from oktopk_amd.config import EngineConfig, OkTopkConfig
from oktopk_amd.trainer import Trainer
from oktopk_amd import AllReducer, Comm, ops

def sync(): torch.cuda.synchronize()

# probe 1: dense trainer + FULL oktopk engine run on a side tensor
cfg = EngineConfig.preset("bert", compressor="dense", dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
eng = AllReducer(Comm(None), EngineConfig(compressor="oktopk", density=0.001,
                 oktopk=OkTopkConfig(dense_warmup_iters=0)))
flat = torch.randn(109_500_000, device="cuda")
for _ in range(3): tr.step()
out=[]
for i in range(12):
    sync(); t0=time.perf_counter()
    tr.step(); eng.run("x", flat)
    sync(); out.append(1000*(time.perf_counter()-t0))
print("dense+engine:", " ".join(f"{x:.1f}" for x in out))

# probe 2: full oktopk trainer, fine-grained phase timing inside the step
cfg2 = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr2 = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg2, dtype="bf16")
for _ in range(3): tr2.step()
opt = tr2.opt
for i in range(9):
    tr2.opt.zero_grad()
    sync(); t0=time.perf_counter()
    loss = tr2._forward_loss()
    sync(); t1=time.perf_counter()
    loss.backward()
    sync(); t2=time.perf_counter()
    opt.flat_grad.copy_(opt.flat_grad_model)
    sync(); t3=time.perf_counter()
    opt.reducer.run("flat", opt.flat_grad)
    sync(); t4=time.perf_counter()
    gn = ops.l2norm(opt.flat_grad)
    sync(); t5=time.perf_counter()
    d = opt.decay_numel; lr=opt.current_lr(); b1,b2=opt.betas
    ops.fused_adam_(opt.flat_param[:d], opt.flat_grad[:d], opt.exp_avg[:d], opt.exp_avg_sq[:d], lr,b1,b2,opt.eps,opt.weight_decay)
    ops.fused_adam_(opt.flat_param[d:], opt.flat_grad[d:], opt.exp_avg[d:], opt.exp_avg_sq[d:], lr,b1,b2,opt.eps,0.0)
    opt.flat_param_model.copy_(opt.flat_param)
    opt.step_count += 1
    sync(); t6=time.perf_counter()
    ms=lambda a,b:1000*(b-a)
    print(f"step{i}: fwd {ms(t0,t1):5.1f} bwd {ms(t1,t2):5.1f} upcast {ms(t2,t3):5.1f} reduce {ms(t3,t4):5.1f} norm {ms(t4,t5):5.1f} adam {ms(t5,t6):5.1f}")

import torch, sys
sys.path.insert(0, "/root/repo")
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer
cfg = EngineConfig.preset("bert", compressor="oktopk", density=0.001, dense_warmup_iters=0)
tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype="bf16")
for it in range(6):
    tr.step()
    st = tr.opt.reducer.states.get("flat")
    g = tr.opt.flat_grad_model
    print(f"it={it} tau_local={st.tau_local:.6e} tau_global={st.tau_global:.6e} "
          f"res_nnz={(st.residual!=0).sum().item()} gnnz={(g!=0).sum().item()}", flush=True)

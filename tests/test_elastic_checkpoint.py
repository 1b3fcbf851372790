"""Elastic shrink + checkpoint/resume tests."""
import os
import torch

from conftest import run_dist


def _shrink_worker(rank):
    import torch.distributed as dist
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    from oktopk_amd.elastic import shrink_comm, apply_shrink

    comm = Comm(dist.group.WORLD)
    cfg = EngineConfig(compressor="oktopk", density=0.05,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(comm, cfg)
    for it in range(3):
        t = torch.randn(2048, generator=torch.Generator().manual_seed(rank * 10 + it))
        eng.run("w", t)
    new_comm = shrink_comm([0])
    if rank == 0:
        apply_shrink(eng, None, new_comm)
        assert eng.comm.size == 1
        t = torch.randn(2048, generator=torch.Generator().manual_seed(99))
        out = eng.run("w", t)
        assert torch.isfinite(out).all()


def test_elastic_shrink_2_to_1():
    run_dist(_shrink_worker, 2)


def test_checkpoint_roundtrip_flat_adam(tmp_path):
    from oktopk_amd.config import EngineConfig, OkTopkConfig
    from oktopk_amd.trainer import Trainer
    from oktopk_amd.utils import save_checkpoint, load_checkpoint

    kw = dict(num_hidden_layers=2, hidden_size=32, num_attention_heads=2,
              intermediate_size=64, vocab_size=300)
    cfg = EngineConfig(compressor="oktopk", density=0.05,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    tr = Trainer("bert_base", batch_size=2, cfg=cfg, model_kwargs=kw, dtype="fp32")
    tr.batches.input_ids.clamp_(max=299); tr.batches.mlm_labels.clamp_(max=299)
    for _ in range(4):
        tr.step()
    path = str(tmp_path / "ck.pth")
    save_checkpoint(path, tr.model, tr.opt, iteration=4, epoch=1)
    res_before = tr.opt.reducer.states["flat"].residual.clone()
    step_before = tr.opt.step_count

    tr2 = Trainer("bert_base", batch_size=2, cfg=cfg, model_kwargs=kw, dtype="fp32")
    it, ep, _ = load_checkpoint(path, tr2.model, tr2.opt)
    assert (it, ep) == (4, 1)
    assert tr2.opt.step_count == step_before
    assert torch.allclose(tr2.opt.reducer.states["flat"].residual, res_before)
    # resumed training continues finite and parameters match the original
    for p1, p2 in zip(tr.model.parameters(), tr2.model.parameters()):
        assert torch.allclose(p1, p2)
    tr2.batches.input_ids.clamp_(max=299); tr2.batches.mlm_labels.clamp_(max=299)
    assert torch.isfinite(torch.tensor(tr2.step()))


def test_checkpoint_roundtrip_sgd(tmp_path):
    from oktopk_amd.config import EngineConfig, OkTopkConfig
    from oktopk_amd.trainer import Trainer
    from oktopk_amd.utils import save_checkpoint, load_checkpoint

    cfg = EngineConfig(compressor="topkA", density=0.05,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    tr = Trainer("vgg16", batch_size=2, cfg=cfg, dtype="fp32")
    for _ in range(2):
        tr.step()
    path = str(tmp_path / "ck.pth")
    save_checkpoint(path, tr.model, tr.opt, iteration=2, epoch=0)
    tr2 = Trainer("vgg16", batch_size=2, cfg=cfg, dtype="fp32")
    it, ep, _ = load_checkpoint(path, tr2.model, tr2.opt)
    for (n1, b1), (n2, b2) in zip(
        sorted(tr.opt.reducer.states.items()), sorted(tr2.opt.reducer.states.items())
    ):
        assert n1 == n2
        assert torch.allclose(b1.residual, b2.residual)

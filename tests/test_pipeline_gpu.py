"""GPU pipeline test: 2-stage BERT pipeline, both ranks on one GPU (gloo
transport for the stage activations, compute on cuda:0) — validates the
GPU compute path of PipelineRuntime."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from conftest import run_dist

CFG = dict(num_hidden_layers=2, hidden_size=64, num_attention_heads=2,
           intermediate_size=128, vocab_size=300, hidden_dropout_prob=0.0,
           attention_probs_dropout_prob=0.0)


def _worker(rank):
    from oktopk_amd.models import bert_base
    from oktopk_amd.pipeline import PipelineRuntime, partition_bert

    torch.cuda.set_device(0)
    torch.manual_seed(0)
    model = bert_base(**CFG)
    stages = partition_bert(model, 2)
    stage = stages[rank].cuda()
    rt = PipelineRuntime(stage, stage_id=rank, num_stages=2,
                         device=torch.device("cuda", 0))
    opt = torch.optim.SGD(stage.parameters(), lr=0.01)
    g = torch.Generator().manual_seed(1)
    mbs = []
    for _ in range(2):
        mb = dict(
            input_ids=torch.randint(0, 300, (2, 16), generator=g).cuda(),
            token_type_ids=torch.zeros(2, 16, dtype=torch.long).cuda(),
            attention_mask=torch.ones(2, 16, dtype=torch.long).cuda(),
            masked_lm_labels=torch.randint(0, 300, (2, 16), generator=g).cuda(),
            next_sentence_label=torch.randint(0, 2, (2,), generator=g).cuda(),
        )
        mbs.append(mb)
    if rank == 0:
        my = [{k: m[k] for k in ("input_ids", "token_type_ids", "attention_mask")}
              for m in mbs]
    else:
        my = [{k: m[k] for k in ("attention_mask", "masked_lm_labels",
                                 "next_sentence_label")} for m in mbs]
    loss = rt.run_step_with_flushes(my, opt)
    if rank == 1:
        assert loss == loss and loss > 0  # finite
    # second step still works after the optimizer update
    loss2 = rt.run_step_with_flushes(my, opt)
    if rank == 1:
        assert loss2 == loss2


def test_pipeline_2stage_on_gpu():
    run_dist(_worker, 2)

"""Model zoo: every registry entry builds and runs forward/backward on CPU."""
import pytest
import torch

from oktopk_amd import models

CIFAR_MODELS = ["vgg11", "vgg16", "resnet20", "resnet56", "alexnet",
                "caffe_cifar", "densenet", "resnext"]


@pytest.mark.parametrize("name", CIFAR_MODELS)
def test_cifar_models_forward_backward(name):
    torch.manual_seed(0)
    m = models.create_net(name)
    x = torch.randn(2, 3, 32, 32)
    out = m(x)
    assert out.shape == (2, 10)
    loss = torch.nn.functional.cross_entropy(out, torch.tensor([1, 3]))
    loss.backward()
    assert all(p.grad is not None for p in m.parameters() if p.requires_grad)


@pytest.mark.parametrize("name", ["resnet18", "resnet50"])
def test_imagenet_models_forward(name):
    m = models.create_net(name)
    out = m(torch.randn(1, 3, 64, 64))
    assert out.shape == (1, 1000)


def test_ptb_lstm():
    m = models.create_net("lstm", vocab_size=200, emb=32, hidden=32)
    x = torch.randint(0, 200, (5, 3))
    logits, hidden = m(x)
    assert logits.shape == (5, 3, 200)
    loss = torch.nn.functional.cross_entropy(
        logits.view(-1, 200), torch.randint(0, 200, (15,))
    )
    loss.backward()


def test_deepspeech_shapes():
    m = models.create_net("lstman4", rnn_hidden_size=64)
    x = torch.randn(2, 1, 161, 101)
    out = m(x)
    assert out.dim() == 3 and out.size(1) == 2 and out.size(2) == 29


def test_deepspeech_param_count_reference_scale():
    """Reference DeepSpeech an4: 5x800 bi-LSTM ~= 27M params
    (LSTM/models/lstman4.py:7)."""
    m = models.create_net("lstman4")
    n = sum(p.numel() for p in m.parameters())
    assert 20e6 < n < 50e6, n


def test_bert_base_param_count():
    """Reference BERT-base: 109.5M params."""
    m = models.create_net("bert_base")
    n = sum(p.numel() for p in m.parameters())
    assert 105e6 < n < 115e6, n


def test_bert_large_param_count():
    m = models.create_net("bert_large")
    n = sum(p.numel() for p in m.parameters())
    assert 300e6 < n < 360e6, n


def test_flops_counter():
    from oktopk_amd.utils import get_model_complexity_info

    m = models.create_net("vgg16")
    flops, params = get_model_complexity_info(m, (3, 32, 32))
    # VGG-16 on 32x32 is ~0.31 GMac; params ~15M
    assert 2e8 < flops < 5e8, flops
    assert 14e6 < params < 16e6, params


def test_sdpa_path_matches_manual_cpu(monkeypatch):
    """OKTOPK_SDPA path == explicit matmul+softmax (p=0, fp32).  The flag is
    read at import; poke the module global directly for the A/B."""
    import oktopk_amd.models.bert as B

    m = B.BertSelfAttention(B.BertConfig(
        hidden_size=64, num_attention_heads=4, attention_probs_dropout_prob=0.0))
    m.eval()
    x = torch.randn(2, 16, 64)
    y_manual = m(x)
    monkeypatch.setattr(B, "_USE_SDPA", True)
    # CPU path ignores SDPA (x.is_cuda gate) — force through by faking cuda
    # is overkill; instead call SDPA directly against the manual output
    q = m.qkv(x).view(2, 16, 3, 4, 16).permute(2, 0, 3, 1, 4)
    ctx = torch.nn.functional.scaled_dot_product_attention(q[0], q[1], q[2])
    y_sdpa = m.out(ctx.transpose(1, 2).reshape(2, 16, 64))
    assert torch.allclose(y_manual, y_sdpa, atol=1e-5), (y_manual - y_sdpa).abs().max()


def test_resnet_variants_forward_backward():
    """preresnet (pre-activation) + resnet_mod (option-A zero-pad shortcut)
    variants (reference preresnet.py / resnet_mod.py)."""
    from oktopk_amd import models

    for name in ("preresnet20", "resnet_mod20"):
        m = models.create_net(name)
        y = m(torch.randn(2, 3, 32, 32))
        assert y.shape == (2, 10)
        y.sum().backward()
        assert all(p.grad is not None for p in m.parameters())
    # option-A shortcut adds no parameters at stage boundaries:
    # resnet_mod20 must have FEWER params than projection-shortcut resnet20
    a = sum(p.numel() for p in models.create_net("resnet_mod20").parameters())
    b = sum(p.numel() for p in models.create_net("resnet20").parameters())
    assert a < b


def test_mnistnet_trains():
    """mnistnet recipe (reference dl_trainer.py:59-76,87) trains end-to-end
    on synthetic MNIST-shape batches."""
    from oktopk_amd.comm import Comm
    from oktopk_amd.trainer import Trainer

    tr = Trainer(model_name="mnistnet", batch_size=4, comm=Comm(None), dtype="fp32")
    losses = [tr.step() for _ in range(3)]
    assert all(torch.isfinite(torch.tensor(losses)))

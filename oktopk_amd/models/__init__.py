"""Model zoo: the three reference workload families, re-implemented.

Reference: VGG/models/ + LSTM/models/ (shared CNN/LSTM zoo, registry at
VGG/models/__init__.py:16-27) and BERT/bert/transformers/modeling.py (HF fork).
These are fresh implementations sized to the reference configs.
"""
from .vgg import VGG, vgg16
from .lstm import DeepSpeech, deepspeech_an4
from .bert import BertConfig, BertForPreTraining, bert_base, bert_large
from .resnet import (
    preresnet20,
    preresnet32,
    preresnet44,
    preresnet56,
    preresnet110,
    resnet_mod20,
    resnet_mod32,
    resnet_mod44,
    resnet_mod56,
    resnet_mod110,
    resnet20, resnet32, resnet44, resnet56, resnet110,
    resnet18, resnet34, resnet50, resnet101, resnet152,
)
from .small_cnns import (
    AlexNet, CaffeCifar, DenseNetCifar, MnistNet, ResNeXtCifar, PTBLSTM,
)

_REGISTRY = {
    # CIFAR CNNs (reference VGG/models registry)
    "vgg11": lambda **kw: VGG("vgg11", **kw),
    "vgg13": lambda **kw: VGG("vgg13", **kw),
    "vgg16": vgg16,
    "vgg19": lambda **kw: VGG("vgg19", **kw),
    "resnet20": resnet20,
    "resnet32": resnet32,
    "resnet44": resnet44,
    "resnet56": resnet56,
    "resnet110": resnet110,
    "preresnet20": preresnet20,
    "preresnet32": preresnet32,
    "preresnet44": preresnet44,
    "preresnet56": preresnet56,
    "preresnet110": preresnet110,
    "resnet_mod20": resnet_mod20,
    "resnet_mod32": resnet_mod32,
    "resnet_mod44": resnet_mod44,
    "resnet_mod56": resnet_mod56,
    "resnet_mod110": resnet_mod110,
    "alexnet": AlexNet,
    "mnistnet": MnistNet,
    "caffe_cifar": CaffeCifar,
    "densenet": DenseNetCifar,
    "resnext": ResNeXtCifar,
    # ImageNet CNNs
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
    # RNN workloads
    "lstm": PTBLSTM,        # PTB word LM (reference models/lstm.py)
    "lstman4": deepspeech_an4,  # DeepSpeech-style AN4 (reference models/lstman4.py)
    # BERT
    "bert_base": bert_base,
    "bert_large": bert_large,
}


def create_net(name: str, **kwargs):
    """Build a model by registry name (reference create_net, VGG/dl_trainer.py:81)."""
    if name not in _REGISTRY:
        raise ValueError(f"unknown model {name!r}; have {sorted(_REGISTRY)}")
    return _REGISTRY[name](**kwargs)

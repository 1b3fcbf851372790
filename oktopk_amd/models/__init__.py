"""Model zoo: the three reference workload families, re-implemented.

Reference: VGG/models/ + LSTM/models/ (shared CNN/LSTM zoo, registry at
VGG/models/__init__.py:16-27) and BERT/bert/transformers/modeling.py (HF fork).
These are fresh implementations sized to the reference configs.
"""
from .vgg import VGG, vgg16
from .lstm import DeepSpeech, deepspeech_an4
from .bert import BertConfig, BertForPreTraining, bert_base, bert_large

_REGISTRY = {
    "vgg16": vgg16,
    "lstman4": deepspeech_an4,
    "bert_base": bert_base,
    "bert_large": bert_large,
}


def create_net(name: str, **kwargs):
    """Build a model by registry name (reference create_net, VGG/dl_trainer.py:81)."""
    if name not in _REGISTRY:
        raise ValueError(f"unknown model {name!r}; have {sorted(_REGISTRY)}")
    return _REGISTRY[name](**kwargs)

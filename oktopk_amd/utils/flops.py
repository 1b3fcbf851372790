"""Per-module FLOPs/params counter via forward hooks.

Capability parity with the vendored ptflops (reference BERT/ptflops/
flops_counter.py:19, used once at startup for per-stage complexity stats,
BERT/bert/main_bert.py:861).  Counts MACs for the module types the reference
workloads use: Conv2d, Linear, LSTM, BatchNorm, LayerNorm, activations,
pooling, Embedding.
"""
from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch
import torch.nn as nn


def _conv2d_flops(m: nn.Conv2d, inp, out) -> int:
    kh, kw = m.kernel_size
    cin = m.in_channels // m.groups
    macs_per_elem = kh * kw * cin
    out_elems = out.numel()
    f = out_elems * macs_per_elem
    if m.bias is not None:
        f += out_elems
    return int(f)


def _linear_flops(m: nn.Linear, inp, out) -> int:
    f = out.numel() * m.in_features
    if m.bias is not None:
        f += out.numel()
    return int(f)


def _lstm_flops(m: nn.LSTM, inp, out) -> int:
    x = inp[0]
    t, b = (x.size(0), x.size(1)) if not m.batch_first else (x.size(1), x.size(0))
    dirs = 2 if m.bidirectional else 1
    total = 0
    isz = m.input_size
    for layer in range(m.num_layers):
        gates = 4 * m.hidden_size * (isz + m.hidden_size + 2)
        total += dirs * t * b * gates
        isz = m.hidden_size * dirs
    return int(total)


def _elemwise(m, inp, out) -> int:
    return int(out.numel() if isinstance(out, torch.Tensor) else 0)


def _embedding_flops(m, inp, out) -> int:
    return 0  # lookups, no MACs (ptflops counts 0 as well)


_HOOKS = {
    nn.Conv2d: _conv2d_flops,
    nn.Linear: _linear_flops,
    nn.LSTM: _lstm_flops,
    nn.BatchNorm1d: _elemwise,
    nn.BatchNorm2d: _elemwise,
    nn.LayerNorm: _elemwise,
    nn.ReLU: _elemwise,
    nn.GELU: _elemwise,
    nn.Hardtanh: _elemwise,
    nn.MaxPool2d: _elemwise,
    nn.AvgPool2d: _elemwise,
    nn.Embedding: _embedding_flops,
}


def get_model_complexity_info(
    model: nn.Module,
    input_res: Tuple = None,
    input_constructor: Optional[Callable] = None,
    print_per_layer_stat: bool = False,
    as_strings: bool = False,
):
    """Returns (flops, params) for one forward pass.

    `input_constructor` may return a dict of kwargs (like ptflops) or a
    tensor; `input_res` builds a float tensor batch of 1 otherwise.
    """
    totals = {"flops": 0}
    handles = []

    def make_hook(fn):
        def hook(mod, inp, out):
            try:
                totals["flops"] += fn(mod, inp, out)
            except Exception:
                pass

        return hook

    for m in model.modules():
        fn = _HOOKS.get(type(m))
        if fn is None:
            # subclasses (e.g. ColsumLinear) count as their base op
            for klass, f in _HOOKS.items():
                if isinstance(m, klass):
                    fn = f
                    break
        if fn is not None:
            handles.append(m.register_forward_hook(make_hook(fn)))

    model.eval()
    with torch.no_grad():
        if input_constructor is not None:
            inp = input_constructor(input_res)
            if isinstance(inp, dict):
                model(**inp)
            else:
                model(inp)
        else:
            model(torch.zeros(1, *input_res))
    for h in handles:
        h.remove()

    params = sum(p.numel() for p in model.parameters())
    flops = totals["flops"]
    if as_strings:
        return f"{flops/1e9:.2f} GMac", f"{params/1e6:.2f} M"
    return flops, params

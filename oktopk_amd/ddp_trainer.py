"""Dense-baseline S-SGD via torch DDP — the alternative training path
(parity with the reference's horovod trainer, VGG/horovod_trainer.py:22
ssgd_with_horovod: wraps the model in the framework-native data-parallel
wrapper instead of the sparse engine)."""
from __future__ import annotations

from typing import Optional

import torch
from torch.nn.parallel import DistributedDataParallel as DDP

from .comm import Comm
from . import models


def ssgd_with_ddp(
    model_name: str = "vgg16",
    batch_size: int = 16,
    lr: float = 0.1,
    comm: Optional[Comm] = None,
    device: Optional[torch.device] = None,
    model_kwargs: Optional[dict] = None,
):
    """Build (ddp_model, optimizer) for dense data-parallel training with
    bucketed, backward-overlapped RCCL allreduce (what horovod gave the
    reference)."""
    comm = comm or Comm(None)
    device = device or (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available()
        else torch.device("cpu")
    )
    model = models.create_net(model_name, **(model_kwargs or {})).to(device)
    if comm.size > 1:
        ids = [device.index] if device.type == "cuda" else None
        model = DDP(model, device_ids=ids)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9, weight_decay=5e-4)
    return model, opt

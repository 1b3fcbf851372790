"""oktopk_amd — MI355X-native sparse-gradient data-parallel training engine.

A ground-up re-design of the capabilities of Shigangli/Ok-Topk (PPoPP'22,
"Near-Optimal Sparse Allreduce for Distributed Deep Learning") for AMD
Instinct MI355X: PyTorch-ROCm frontend, hand-written HIP/CDNA4 kernels for
every selection/compaction/merge/optimizer hot op, and RCCL collectives over
xGMI instead of mpi4py on CPU-staged buffers.
"""

__version__ = "0.1.0"

from .config import EngineConfig, OkTopkConfig  # noqa: F401
from .comm import Comm, init_from_env  # noqa: F401
from .allreducer import AllReducer, COMPRESSORS  # noqa: F401


def __getattr__(name):
    # Lazy top-level exports for the heavier user-facing classes (keeps
    # `import oktopk_amd` light for kernels-only use; no circular imports).
    if name in ("DistributedOptimizer", "FlatBertAdam"):
        from . import optimizer

        return getattr(optimizer, name)
    if name == "Trainer":
        from .trainer import Trainer

        return Trainer
    raise AttributeError(name)

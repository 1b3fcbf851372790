from .metrics import accuracy_topk, wer, cer, perplexity, GreedyDecoder  # noqa: F401
from .logging import get_logger, MetricWriter  # noqa: F401
from .comm_model import alpha_beta_time, predict_allreduce_time  # noqa: F401
from .checkpoint import save_checkpoint, load_checkpoint  # noqa: F401
from .flops import get_model_complexity_info  # noqa: F401

from .tokenization import WordPieceTokenizer, BasicTokenizer  # noqa: F401
from .ptb import PTBReader, ptb_batchify  # noqa: F401
from .bert_dataset import BertPretrainingDataset  # noqa: F401
from .audio import SpectrogramDataset, compute_spectrogram  # noqa: F401
from .vision import cifar_like_dataset, Hdf5ImagenetDataset  # noqa: F401

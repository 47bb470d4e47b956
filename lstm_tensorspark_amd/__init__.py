"""MI355X-native distributed LSTM training engine.

A from-scratch replacement for EmanuelOverflow/LSTM-TensorSpark's
capabilities (SURVEY.md §8 capability contract): train()/fit() API,
multi-layer/bidirectional LSTM, truncated BPTT, dropout, grad clipping,
LM head, bf16 compute with fp32 master weights, synchronous data-parallel
training over RCCL/xGMI, hand-written CDNA4 HIP kernels for the cell
forward/backward hot path.
"""
__version__ = "0.1.0"

from .config import Config, get_preset, PRESETS  # noqa: F401
from .trainer import Trainer  # noqa: F401
from .models.lstm import LSTMModel  # noqa: F401
from .checkpoint import (  # noqa: F401
    save_checkpoint, load_checkpoint,
    export_weights_safetensors, load_weights_safetensors,
)
from .serving import ContinuousBatcher  # noqa: F401

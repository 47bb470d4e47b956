from .lstm import LSTMModel  # noqa: F401

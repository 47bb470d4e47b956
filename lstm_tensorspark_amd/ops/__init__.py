from . import extension, functional  # noqa: F401
from .lstm_layer import lstm_layer  # noqa: F401

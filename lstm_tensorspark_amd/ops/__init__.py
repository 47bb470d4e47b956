"""Compute ops: HIP extension loader, reference semantics, layer op.

NOTE: the `lstm_layer` FUNCTION is intentionally NOT re-exported here --
it would shadow the `ops.lstm_layer` MODULE on `from ..ops import
lstm_layer`, hiding enable_side_dw()/wait_side_stream().  Import the
function from ops.lstm_layer directly.
"""
from . import extension, functional, lstm_layer  # noqa: F401

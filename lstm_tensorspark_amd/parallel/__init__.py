from .ddp import GradBucketer, broadcast_parameters  # noqa: F401
from .launch import DistContext, init_from_env, cleanup, run_spawned  # noqa: F401

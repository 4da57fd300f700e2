from .ddp import Reducer, init_distributed_from_env  # noqa: F401

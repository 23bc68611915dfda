from .ddp import GradAllReduce, init_distributed  # noqa: F401

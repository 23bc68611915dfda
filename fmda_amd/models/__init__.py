from .bigru import BiGRU  # noqa: F401
from .checkpoint import load_checkpoint, save_checkpoint  # noqa: F401

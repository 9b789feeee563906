from .checkpoint import save_checkpoint, load_checkpoint  # noqa: F401
from .trace import Tracer  # noqa: F401

from . import state  # noqa: F401
from .state import initialize_model_parallel  # noqa: F401

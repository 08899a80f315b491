from .net import Net  # noqa: F401
from .mlp import MLP  # noqa: F401

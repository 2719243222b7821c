from .arena import ParamArena, SGD, Adam  # noqa: F401
from . import layers  # noqa: F401

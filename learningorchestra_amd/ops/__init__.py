from ._ext import ext, has_ext, require_ext  # noqa: F401
from . import functional  # noqa: F401

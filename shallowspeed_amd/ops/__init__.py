from . import functional  # noqa: F401
from ._ext import build_ext, load_ext  # noqa: F401

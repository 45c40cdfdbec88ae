from . import checkpoint, logger, patch, utils  # noqa: F401
from .logger import logger as _logger  # noqa: F401

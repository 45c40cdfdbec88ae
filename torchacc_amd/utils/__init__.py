from . import (checkpoint, consolidate_and_reshard_ckpts, cpu_offload,  # noqa: F401
               decompose, import_utils, logger, patch, trace, utils)
from .logger import logger as _logger  # noqa: F401

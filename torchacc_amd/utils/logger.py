"""Framework logger.

Mirrors reference behavior (torchacc/utils/logger.py): a single `logging` logger
whose level is taken from the ``ACC_LOG_LEVEL`` environment variable.
"""
import logging
import os

logger = logging.getLogger("TorchAccAMD")
if not logger.handlers:
    _h = logging.StreamHandler()
    _h.setFormatter(
        logging.Formatter("[%(asctime)s %(levelname)s %(name)s] %(message)s"))
    logger.addHandler(_h)
logger.setLevel(os.environ.get("ACC_LOG_LEVEL", "WARNING").upper())
logger.propagate = False

"""Shared ``"ActiveLearning"`` logger. Reference: src/utils/setup_logging.py:6-30."""

import logging
import os

LOGGER_NAME = "ActiveLearning"

_FMT = "%(asctime)s.%(msecs)03d %(levelname)s %(name)s: %(message)s"
_DATEFMT = "%Y-%m-%d %H:%M:%S"


def setup_logging(log_dir: str, log_filename: str, level=logging.INFO) -> logging.Logger:
    """Configure the shared logger with a file handler + console handler."""
    os.makedirs(log_dir, exist_ok=True)
    logger = logging.getLogger(LOGGER_NAME)
    logger.setLevel(level)
    logger.handlers.clear()
    formatter = logging.Formatter(_FMT, datefmt=_DATEFMT)

    fh = logging.FileHandler(os.path.join(log_dir, log_filename))
    fh.setFormatter(formatter)
    logger.addHandler(fh)

    ch = logging.StreamHandler()
    ch.setFormatter(formatter)
    logger.addHandler(ch)
    logger.propagate = False
    return logger


def get_logger() -> logging.Logger:
    return logging.getLogger(LOGGER_NAME)

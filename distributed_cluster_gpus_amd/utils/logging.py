"""Rotating project.log, format-compatible with the reference
(simcore/logger_config.py:7-32): "%(asctime)s | %(levelname)-8s | SIMU_DC |
%(message)s", 5 MB x 3 backups, DEBUG level."""
import logging
import os
from logging.handlers import RotatingFileHandler

LOGGER_NAME = "SIMU_DC"


def get_logger(log_dir: str) -> logging.Logger:
    os.makedirs(log_dir, exist_ok=True)
    log_path = os.path.abspath(os.path.join(log_dir, "project.log"))
    logger = logging.getLogger(LOGGER_NAME)
    logger.setLevel(logging.DEBUG)
    logger.propagate = False
    # idempotent per target file (the reference guards on hasHandlers(), which
    # breaks under a pre-configured root logger and when one process runs
    # several simulations; key on the actual file instead).
    for h in logger.handlers:
        if isinstance(h, RotatingFileHandler) and h.baseFilename == log_path:
            return logger
    fh = RotatingFileHandler(log_path, mode="a", maxBytes=5_000_000,
                             backupCount=3, encoding="utf-8")
    fh.setFormatter(logging.Formatter(
        "%(asctime)s | %(levelname)-8s | %(name)s | %(message)s"))
    logger.addHandler(fh)
    return logger

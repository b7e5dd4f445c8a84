"""Rotating project.log, format-compatible with the reference
(simcore/logger_config.py:7-32): "%(asctime)s | %(levelname)-8s | SIMU_DC |
%(message)s", 5 MB x 3 backups, DEBUG level."""
import logging
import os
from logging.handlers import RotatingFileHandler

LOGGER_NAME = "SIMU_DC"


def get_logger(log_dir: str) -> logging.Logger:
    os.makedirs(log_dir, exist_ok=True)
    log_path = os.path.join(log_dir, "project.log")
    logger = logging.getLogger(LOGGER_NAME)
    if not logger.hasHandlers():
        logger.setLevel(logging.DEBUG)
        fh = RotatingFileHandler(log_path, mode="a", maxBytes=5_000_000,
                                 backupCount=3, encoding="utf-8")
        fh.setFormatter(logging.Formatter(
            "%(asctime)s | %(levelname)-8s | %(name)s | %(message)s"))
        logger.addHandler(fh)
    return logger

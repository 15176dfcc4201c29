"""Fail-open internal error sink: a file-backed logger with the [TraceML-AMD]
prefix, never propagating to the user's handlers
(reference: loggers/error_log.py:115)."""

from __future__ import annotations

import logging
import os
from typing import Optional

_logger: Optional[logging.Logger] = None


def setup_error_logger(session_dir: Optional[str] = None) -> logging.Logger:
    global _logger
    if _logger is not None:
        return _logger
    logger = logging.getLogger("traceml_amd.errors")
    logger.setLevel(logging.WARNING)
    logger.propagate = False
    if session_dir:
        try:
            os.makedirs(session_dir, exist_ok=True)
            handler = logging.FileHandler(
                os.path.join(session_dir, "traceml_errors.log")
            )
            handler.setFormatter(
                logging.Formatter("[TraceML-AMD] %(asctime)s %(levelname)s %(message)s")
            )
            logger.addHandler(handler)
        except OSError:
            logger.addHandler(logging.NullHandler())
    else:
        logger.addHandler(logging.NullHandler())
    _logger = logger
    return logger


def get_error_logger() -> logging.Logger:
    return _logger if _logger is not None else setup_error_logger()

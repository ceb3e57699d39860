"""Shared stdout logging.

The reference copy-pastes an identical ``configure_logger`` into all four
stages (``stage_1_train_model.py:145-158`` = ``stage_2:83-96`` =
``stage_3:64-77`` = ``stage_4:137-150``); this module is the single shared
implementation, byte-compatible in its record format:

    ``%(asctime)s - %(levelname)s - %(module)s.%(funcName)s - %(message)s``

The global level defaults to INFO and is settable from the pipeline config
(``logging.log_level`` in the yaml — reference ``bodywork.yaml:83-84``).
"""
from __future__ import annotations

import logging
import sys

LOG_FORMAT = (
    "%(asctime)s - "
    "%(levelname)s - "
    "%(module)s.%(funcName)s - "
    "%(message)s"
)

_CONFIGURED: set = set()


def configure_logger(
    name: str = "bodywork_mlops_demo_amd", level: str | int = logging.INFO
) -> logging.Logger:
    """Return a logger writing to stdout in the reference record format."""
    log = logging.getLogger(name)
    if isinstance(level, str):
        level = getattr(logging, level.upper(), logging.INFO)
    if name not in _CONFIGURED:
        handler = logging.StreamHandler(sys.stdout)
        handler.setFormatter(logging.Formatter(LOG_FORMAT))
        log.addHandler(handler)
        _CONFIGURED.add(name)
    log.setLevel(level)
    return log


def set_global_log_level(level: str | int) -> None:
    """Apply the pipeline-config log level to every framework logger."""
    configure_logger(level=level)
    for name in list(_CONFIGURED):
        configure_logger(name, level=level)

"""Logger setup (reference logging.py clones vLLM's dict config; here it is
our own single-format config)."""

from __future__ import annotations

import logging
import logging.config

DEFAULT_LOGGER_NAME = __name__.split(".")[0]

_CONFIG = {
    "version": 1,
    "disable_existing_loggers": False,
    "formatters": {
        DEFAULT_LOGGER_NAME: {
            "format": (
                "%(levelname)s %(asctime)s.%(msecs)03d %(name)s:%(lineno)d] %(message)s"
            ),
            "datefmt": "%m-%d %H:%M:%S",
        }
    },
    "handlers": {
        DEFAULT_LOGGER_NAME: {
            "class": "logging.StreamHandler",
            "formatter": DEFAULT_LOGGER_NAME,
            "stream": "ext://sys.stdout",
        }
    },
    "loggers": {
        DEFAULT_LOGGER_NAME: {
            "handlers": [DEFAULT_LOGGER_NAME],
            "level": "INFO",
            "propagate": False,
        }
    },
}

logging.config.dictConfig(_CONFIG)


def init_logger(name: str) -> logging.Logger:
    if not name.startswith(DEFAULT_LOGGER_NAME):
        name = f"{DEFAULT_LOGGER_NAME}.{name}"
    return logging.getLogger(name)

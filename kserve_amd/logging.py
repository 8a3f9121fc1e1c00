"""Logger configuration (reference parity: python/kserve/kserve/logging.py)."""

import logging
import logging.config

KSERVE_AMD_LOGGER_NAME = "kserve_amd"
TRACE_LOGGER_NAME = "kserve_amd.trace"

KSERVE_AMD_LOG_CONFIG = {
    "version": 1,
    "disable_existing_loggers": False,
    "formatters": {
        "default": {
            "()": "logging.Formatter",
            "fmt": "%(asctime)s.%(msecs)03d %(process)s %(name)s "
            "%(levelname)s [%(funcName)s():%(lineno)s] %(message)s",
            "datefmt": "%Y-%m-%d %H:%M:%S",
        },
    },
    "handlers": {
        "default": {
            "formatter": "default",
            "class": "logging.StreamHandler",
            "stream": "ext://sys.stderr",
        },
    },
    "loggers": {
        KSERVE_AMD_LOGGER_NAME: {
            "handlers": ["default"],
            "level": "INFO",
            "propagate": False,
        },
        TRACE_LOGGER_NAME: {
            "handlers": ["default"],
            "level": "INFO",
            "propagate": False,
        },
        "uvicorn": {"handlers": ["default"], "level": "INFO", "propagate": False},
        "uvicorn.error": {"handlers": ["default"], "level": "INFO", "propagate": False},
        "uvicorn.access": {"handlers": ["default"], "level": "INFO", "propagate": False},
    },
}

logger = logging.getLogger(KSERVE_AMD_LOGGER_NAME)
trace_logger = logging.getLogger(TRACE_LOGGER_NAME)

_configured = False


def configure_logging(log_config=None):
    global _configured
    logging.config.dictConfig(log_config or KSERVE_AMD_LOG_CONFIG)
    _configured = True

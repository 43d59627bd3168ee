"""Framework logger configuration.

Reference capability (pylzy/lzy/logs/config.py:32-66): a `lzy` logger
hierarchy configured from LZY_LOG_LEVEL / LZY_LOG_CONFIG_PATH with a
colored console formatter, plus get_logger / get_remote_logger
accessors.  Here "remote" loggers are the worker ranks' — their records
ride the live op-log stream (utils/logs.py) back to the driver console,
so both hierarchies print locally with the same format.
"""
from __future__ import annotations

import logging
import logging.config
import os
import sys
from typing import Any, Dict, Optional

LZY_LOG_LEVEL = "LZY_LOG_LEVEL"
LZY_LOG_CONFIG_PATH = "LZY_LOG_CONFIG_PATH"

_DEFAULT_FORMAT = "[LZY] %(asctime)s %(levelname)-7s %(name)s: %(message)s"


def get_logging_config() -> Dict[str, Any]:
    """Level from LZY_LOG_LEVEL (default INFO); a yaml/json dictConfig
    file from LZY_LOG_CONFIG_PATH overrides everything."""
    path = os.environ.get(LZY_LOG_CONFIG_PATH)
    if path:
        import json

        with open(path) as f:
            if path.endswith((".yaml", ".yml")):
                import yaml

                return yaml.safe_load(f)
            return json.load(f)
    level = os.environ.get(LZY_LOG_LEVEL, "INFO").upper()
    return {
        "version": 1,
        "disable_existing_loggers": False,
        "formatters": {"lzy": {"format": _DEFAULT_FORMAT}},
        "handlers": {
            "lzy_console": {
                "class": "logging.StreamHandler",
                "formatter": "lzy",
                "stream": "ext://sys.stderr",
            }
        },
        "loggers": {
            "lzy": {"level": level, "handlers": ["lzy_console"],
                    "propagate": False},
            "remote": {"level": level, "handlers": ["lzy_console"],
                       "propagate": False},
        },
    }


def configure_logging(config: Optional[Dict[str, Any]] = None) -> None:
    logging.config.dictConfig(config if config is not None
                              else get_logging_config())


def get_logger(name: str) -> logging.Logger:
    parent = logging.getLogger("lzy")
    if not name.startswith(parent.name + "."):
        return parent.getChild(name)
    return logging.getLogger(name)


def get_remote_logger(name: str) -> logging.Logger:
    """Logger for code running on worker ranks — records are captured by
    the op-log router and stream to the driver while the op RUNS."""
    return logging.getLogger("remote").getChild(name)

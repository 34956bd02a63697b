"""Logging factory (reference parity: python_client/kubetorch/logger.py).
Structured key=value console logs; JSON when KT_LOG_JSON=1 (pod side)."""
import json
import logging
import os
import sys
import time


class _JsonFormatter(logging.Formatter):
    def format(self, record):
        out = {
            "ts": round(time.time(), 3),
            "level": record.levelname,
            "logger": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        return json.dumps(out)


def get_logger(name="kubetorch_amd", level=None):
    logger = logging.getLogger(name)
    if logger.handlers:
        return logger
    handler = logging.StreamHandler(sys.stderr)
    if os.environ.get("KT_LOG_JSON") == "1":
        handler.setFormatter(_JsonFormatter())
    else:
        handler.setFormatter(logging.Formatter(
            "[%(asctime)s %(levelname)s %(name)s] %(message)s", "%H:%M:%S"))
    logger.addHandler(handler)
    logger.setLevel(level or os.environ.get("KT_LOG_LEVEL", "INFO").upper())
    logger.propagate = False
    return logger


logger = get_logger()

"""Structured logging setup.

Parity target: the reference's JSON console + rotating-file logging
(/root/reference/llm_gateway_core/utils/logging_setup.py:14-54) without the
python-json-logger dependency — a small stdlib JSON formatter instead.
"""

from __future__ import annotations

import json
import logging
import logging.config
import os
from datetime import datetime, timezone


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        payload = {
            "asctime": datetime.fromtimestamp(record.created, tz=timezone.utc).isoformat(),
            "levelname": record.levelname,
            "name": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info:
            payload["exc_info"] = self.formatException(record.exc_info)
        for key in ("request_id",):
            v = getattr(record, key, None)
            if v is not None:
                payload[key] = v
        return json.dumps(payload, ensure_ascii=False)


def configure_logging(log_dir: str = "logs", level: str = "INFO") -> None:
    os.makedirs(log_dir, exist_ok=True)
    logging.config.dictConfig(
        {
            "version": 1,
            "disable_existing_loggers": False,
            "formatters": {
                "json": {"()": JsonFormatter},
                "plain": {"format": "%(asctime)s %(levelname)s %(name)s %(message)s"},
            },
            "handlers": {
                "console": {
                    "class": "logging.StreamHandler",
                    "formatter": "plain",
                    "level": level,
                },
                "file": {
                    "class": "logging.handlers.RotatingFileHandler",
                    "formatter": "json",
                    "filename": os.path.join(log_dir, "gateway.log"),
                    "maxBytes": 256 * 1024,
                    "backupCount": 5,
                    "level": level,
                },
            },
            "root": {"handlers": ["console", "file"], "level": level},
            "loggers": {
                # demote noisy HTTP internals, as the reference does (logging_setup.py:47-52)
                "httpcore": {"level": "WARNING"},
                "httpx": {"level": "WARNING"},
                "uvicorn.access": {"level": "WARNING"},
            },
        }
    )

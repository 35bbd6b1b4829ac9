"""Run the gateway: ``python main.py`` (parity: /root/reference/main.py:119-127)."""

from __future__ import annotations

import logging
import sys

import uvicorn

from llmapigateway_amd.config.loader import ConfigError
from llmapigateway_amd.config.settings import Settings
from llmapigateway_amd.gateway.app import create_app
from llmapigateway_amd.utils.logging_setup import configure_logging


def main() -> None:
    configure_logging()
    settings = Settings.from_env()
    try:
        app = create_app(settings=settings)
    except ConfigError as e:
        # initial-load failure is fatal with a clean message, matching the
        # reference's policy (loader.py:74,100,164); editor reloads stay
        # non-fatal
        logging.getLogger("llmapigateway").error("Configuration error: %s", e)
        sys.exit(1)
    uvicorn.run(app, host=settings.gateway_host, port=settings.gateway_port, log_level="info")


if __name__ == "__main__":
    main()

from .logging_setup import configure_logging

__all__ = ["configure_logging"]

from .rotation import ModelRotationDB
from .usage import TokensUsageDB

__all__ = ["ModelRotationDB", "TokensUsageDB"]

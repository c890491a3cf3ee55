from .config import LogRegConfig
from .logreg import LogReg

__all__ = ["LogRegConfig", "LogReg"]

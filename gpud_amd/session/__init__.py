from .session import Session

__all__ = ["Session"]

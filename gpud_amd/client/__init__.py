from .v1 import Client

__all__ = ["Client"]

from .app import create_app
from .server import Server, generate_self_signed_cert

__all__ = ["create_app", "Server", "generate_self_signed_cert"]

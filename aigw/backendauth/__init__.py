from .auth import AuthHandler, build_auth_handler
from .sigv4 import sign_sigv4

__all__ = ["AuthHandler", "build_auth_handler", "sign_sigv4"]

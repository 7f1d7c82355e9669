from .proxy import MCPProxy
from .session import SessionCrypto

__all__ = ["MCPProxy", "SessionCrypto"]

from .server import GatewayServer

__all__ = ["GatewayServer"]

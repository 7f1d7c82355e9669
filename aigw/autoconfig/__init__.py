from .autoconfig import config_from_env

__all__ = ["config_from_env"]

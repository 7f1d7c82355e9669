from .services import GPUServices, extract_chat_text

__all__ = ["GPUServices", "extract_chat_text"]

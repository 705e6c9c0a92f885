from .logging import configure, get_logger, ring_buffer

__all__ = ["configure", "get_logger", "ring_buffer"]

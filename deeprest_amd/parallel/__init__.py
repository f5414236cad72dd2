from .dist import DistContext, init_distributed, is_distributed

__all__ = ["DistContext", "init_distributed", "is_distributed"]

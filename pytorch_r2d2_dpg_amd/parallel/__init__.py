from .dist import DistContext

__all__ = ["DistContext"]

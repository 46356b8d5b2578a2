from .lazy_init import LazyInitContext

__all__ = ["LazyInitContext"]

from .dist import DistContext, init_distributed  # noqa: F401

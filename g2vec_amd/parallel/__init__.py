from .dist import DistContext, init_dist  # noqa: F401

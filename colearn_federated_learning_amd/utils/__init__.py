from .dist import init_distributed, local_device, rank_world  # noqa: F401

from .dist import init_distributed, get_rank, get_world_size, barrier

__all__ = ["init_distributed", "get_rank", "get_world_size", "barrier"]

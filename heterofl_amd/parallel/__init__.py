from .dist import DistContext, init_distributed, distributed_combine

__all__ = ['DistContext', 'init_distributed', 'distributed_combine']

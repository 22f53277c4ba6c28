from .runtime import FedRuntime, get_runtime, init_runtime, local_rank, rank, set_runtime, size

__all__ = ["FedRuntime", "get_runtime", "init_runtime", "local_rank", "rank", "set_runtime", "size"]

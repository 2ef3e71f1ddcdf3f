from .shard import Comm, get_comm, init_from_env

__all__ = ["Comm", "get_comm", "init_from_env"]

from .comm import Comm, init_from_env

__all__ = ["Comm", "init_from_env"]

from .comm import Communicator, init_from_env

__all__ = ["Communicator", "init_from_env"]

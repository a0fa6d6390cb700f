from .comm import Communicator, get_communicator, init_parallel, init_tp

__all__ = ["Communicator", "get_communicator", "init_parallel", "init_tp"]

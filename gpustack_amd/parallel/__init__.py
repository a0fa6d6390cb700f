from .comm import Communicator, get_communicator, init_parallel, init_tp
from .cp import CPMeta, build_cp_prefill, cp_bounds

__all__ = ["Communicator", "get_communicator", "init_parallel", "init_tp",
           "CPMeta", "build_cp_prefill", "cp_bounds"]

from .comm import Comm, SingleComm, DistComm, init_dist_comm  # noqa: F401

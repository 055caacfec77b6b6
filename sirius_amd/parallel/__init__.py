"""Distributed communication environment.

Reference behavior: src/core/mpi (mpi::Communicator, communicator.hpp:288;
grid splits simulation_context.cpp:1301-1334).

MI355X-native equivalent: torch.distributed process groups — backend
"nccl" (RCCL over xGMI) with one process per GPU, "gloo" for CPU test
runs. k-point data parallelism uses the world group; density/dm
accumulation is an all-reduce; band-energy sync an all-gather
(reference call-sites: density.cpp:1336/:1348, k_point_set.cpp:18-44).
"""

from .comm import Comm, get_comm, init_distributed  # noqa: F401

from .comms import (allreduce_sum, comm_rank, comm_size, grid_dims,
                    init_comms, is_distributed)

__all__ = ["allreduce_sum", "comm_rank", "comm_size", "grid_dims",
           "init_comms", "is_distributed"]

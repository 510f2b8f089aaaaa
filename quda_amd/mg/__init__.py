from .transfer import Transfer, block_orthonormalize, generate_null_vectors
from .coarse import CoarseOp, build_coarse_op
from .mg import MG, MGParam

__all__ = ["Transfer", "block_orthonormalize", "generate_null_vectors",
           "CoarseOp", "build_coarse_op", "MG", "MGParam"]

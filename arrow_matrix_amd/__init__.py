"""arrow_matrix_amd — MI355X-native arrow-decomposition iterated-SpMM engine.

A from-scratch rebuild of the hot path of spcl/arrow-matrix (PPoPP'24
"Arrow Matrix Decomposition"): the `spmm_arrow` entry point and the
`ArrowDecompositionMPI` / `ArrowSlimMPI` / `MatrixSlice` API, with the
compute done by a hand-written CDNA4 HIP kernel behind a C ABI
(include/arrow_spmm.h) and the communication by RCCL over xGMI
(torch.distributed). See DESIGN.md.
"""
from .arrow_matrix import ArrowMatrix          # noqa: F401
from .arrow_slim import ArrowSlimMPI           # noqa: F401
from .arrow_dec import ArrowDecompositionMPI   # noqa: F401
from .comm import Comm, TorchDistComm, default_comm  # noqa: F401

__version__ = "0.1.0"

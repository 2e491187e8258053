"""Utilities mirroring arrow/common/utils.py (str2bool:9-17, mpi_print:58-60,
generate_sparse_matrix:63-87, generate_dense_matrix:90-99)."""
import argparse
from typing import Union

import numpy as np
from scipy import sparse


def str2bool(v: Union[str, bool]) -> bool:
    if isinstance(v, bool):
        return v
    if v.lower() in ('yes', 'true', 't', 'y', '1'):
        return True
    if v.lower() in ('no', 'false', 'f', 'n', '0'):
        return False
    raise argparse.ArgumentTypeError('Boolean value expected.')


def relabel_nodes(g, mapping):
    """Relabel graph nodes by a mapping (reference utils.py:20-51):
    g' = I[order] @ g @ I[order]^T."""
    from scipy import sparse as sp
    if not isinstance(g, (sp.csr_array, sp.csr_matrix)):
        raise TypeError("The graph must be a SciPy-compatible CSR array or matrix.")
    if g.shape[0] != g.shape[1]:
        raise ValueError("The matrix must be square.")
    labels = list(range(g.shape[0]))
    if sorted(mapping.keys()) != labels:
        raise ValueError("The keys of the mapping must be the rows of the "
                         "graph's matrix representation.")
    if sorted(mapping.values()) != labels:
        raise ValueError("The values of the mapping must be the rows of the "
                         "graph's matrix representation.")
    order = [mapping[i] for i in range(g.shape[0])]
    I = sp.eye(g.shape[0], format='coo', dtype=np.int32)
    I.row = I.row[order]
    I = I.tocsr()
    return I @ g @ I.T


def time_to_ms(runtime: float) -> int:
    """Reference utils.py:54-55."""
    return int(runtime * 1000)


def mpi_print(rank: int, msg: str):
    if rank == 0:
        print(msg, flush=True)


def generate_sparse_matrix(rows: int, cols: int, nnz: int, dtype,
                           rng: np.random.Generator) -> sparse.csr_matrix:
    """Fixed nonzeros-per-row random CSR (reference utils.py:63-87: nnz may
    round up to a multiple of rows; duplicates summed)."""
    nnzpr = int(np.ceil(nnz / rows))
    actual_nnz = nnzpr * rows
    data = rng.random((actual_nnz,), dtype=dtype)
    indptr = np.arange(0, actual_nnz + 1, nnzpr, dtype=np.int64)
    indices = rng.integers(0, cols, size=(actual_nnz,), dtype=np.int64)
    tmp = sparse.csr_matrix((data, indices, indptr), shape=(rows, cols), dtype=dtype)
    tmp.sum_duplicates()
    tmp.sort_indices()
    return tmp


def generate_dense_matrix(rows: int, cols: int, dtype, rng: np.random.Generator) -> np.ndarray:
    return 2 * rng.random((rows, cols), dtype=dtype) - 1

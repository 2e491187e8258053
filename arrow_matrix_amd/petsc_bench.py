"""benchmark_spmm for the PETSc-style path — reference signature kept
(arrow/baseline/spmm_petsc.py:398-495): load-or-generate a 1D row slice,
build the MatrixSlice tables, iterate SpMM with per-phase timers and the
fail-allreduce abort semantics."""
import time
from typing import Optional

import numpy as np
import torch
from scipy import sparse

from .comm import default_comm
from .common import utils, wb_logging
from .matrix_slice import MatrixSlice
from .spmm_petsc import SpmmPETSc


def load_matrix_slice(some_slice: str, rank: int):
    """`{name}.part.{x}.slice.{y}.npz` naming: replace the second-to-last
    dot component with this rank (reference spmm_petsc.py:82-102)."""
    parts = some_slice.split('.')
    parts[-2] = str(rank)
    return sparse.load_npz('.'.join(parts)).tocsr()


def benchmark_spmm(matrix_slice_file: Optional[str], k: int, iterations: int,
                   device: str, wandb_api_key=None, dtype=np.float32,
                   rng: Optional[np.random.Generator] = None,
                   gpu_tiling: bool = False, dryrun: bool = False,
                   mem_fraction: float = 0.9):
    if gpu_tiling:
        raise NotImplementedError(
            "GPU column tiling (spmm_petsc.py:323-395) is unnecessary with "
            "288 GB HBM3E per GPU and is not reproduced (DESIGN.md §next)")
    if np.dtype(dtype) != np.float32 and device != 'cpu':
        raise NotImplementedError(
            "the HIP kernels compute in fp32 (the reference benchmark "
            "default, arrow_bench.py:21); float64 runs with --device cpu")
    rng = rng if rng is not None else np.random.default_rng(42)
    comm = default_comm()
    name = "PETSc_v0.1_AMD"
    dataset_name = (matrix_slice_file.split('.')[0]
                    if matrix_slice_file is not None else None)
    wb_logging.wandb_init(comm, dataset_name, k, iterations, device, name, 0,
                          wandb_api_key)

    if matrix_slice_file is None:
        scale = 4 * 1024
        A_i = utils.generate_sparse_matrix(scale, comm.size * scale,
                                           scale * 10, dtype, rng)
    else:
        nr_parts = int(matrix_slice_file.split('.')[-4])
        if nr_parts != comm.size:
            raise ValueError(f"Number of parts in file name ({nr_parts}) does "
                             f"not match number of ranks ({comm.size})")
        A_i = load_matrix_slice(matrix_slice_file, comm.rank).astype(dtype)
        A_i.eliminate_zeros()
        A_i.sort_indices()
        A_i.sum_duplicates()

    if dryrun:
        return None

    ms = MatrixSlice.initialize(comm, sparse.csr_matrix(A_i))
    engine = SpmmPETSc(comm, ms, device=device)

    Y = None
    for i in range(iterations):
        X_local = 2 * rng.random((A_i.shape[0], k), dtype=dtype) - 1
        fail = False
        try:
            wb_logging.set_iteration_data({"iteration": i})
            tic = time.perf_counter()
            Y = engine.spmm(X_local)
            toc = time.perf_counter()
            wb_logging.log({"spmm_time": toc - tic})
            if comm.rank == 0:
                print("RANK", comm.rank, "Iteration", i, " -- ", toc - tic,
                      "s", flush=True)
        except Exception as e:
            print("RANK", comm.rank, "EXCEPTION", e, flush=True)
            fail = True
        flag = torch.tensor([1 if fail else 0], dtype=torch.int64)
        comm.allreduce_max_(flag)
        if int(flag.item()):
            print("RANK", comm.rank, "FAILED", flush=True)
            break

    wb_logging.finish()
    comm.barrier()
    return Y

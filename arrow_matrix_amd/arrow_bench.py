"""bench_spmm — the reference benchmark driver, kept signature.

Mirrors arrow/arrow_bench.py:12-137: generate-or-load a decomposition,
initialize the engine, iterate `step()` with per-iteration timers and the
fail-allreduce abort semantics (arrow_bench.py:121-134). The synthetic
fallback replaces igraph Barabási + arrow_decomposition (arrow_bench.py:28-41)
with the numpy-native synth generator (same on-disk format)."""
import os
import sys
import time
from typing import Union

import numpy as np
import torch

from . import graphio, synth
from .arrow_dec import ArrowDecompositionMPI
from .comm import default_comm
from .common import wb_logging


def bench_spmm(path: Union[str, None],
               width: int,
               n_features: int,
               iterations: int,
               blocked: bool,
               device: str,
               p_per_side: int = 3,
               ba_neighbors: int = 5,
               wandb_api_key: str = None,
               datatype=np.float32,
               slim: bool = True,
               npy_format: bool = True):
    assert width > 0
    assert not slim or blocked  # reference arrow_dec_mpi.py:131
    comm = default_comm()

    if path is None:
        path = 'tmp/test_ba' + "_" + str(p_per_side) + "_" + str(ba_neighbors)
        if comm.rank == 0:
            os.makedirs("tmp", exist_ok=True)
            decomp = synth.synth_arrow_decomposition(
                width, [p_per_side], avg_deg=ba_neighbors, seed=503,
                block_diagonal=blocked)
            if npy_format:
                graphio.save_decomposition_new(decomp, path, width,
                                               block_diagonal=blocked)
            else:
                graphio.save_decomposition(decomp, path, width,
                                           block_diagonal=blocked)
            print("DATASET GENERATED -- ", p_per_side * width, " vertices", flush=True)
        comm.barrier()

    name = "Arrow_v0.45_BlockDiagonal_Slim_AMD"
    wb_logging.wandb_init(comm, path, n_features, iterations, device, name,
                          width, wandb_api_key)

    blocks, n_blocks, to_prev, to_next = ArrowDecompositionMPI.load_decomposition_new(
        comm, path, width, is_block_diagonal=blocked, datatype=datatype,
        slim=slim, use_npy=npy_format, use_mmap=False)

    comm.barrier()
    if np.sum(n_blocks) == 0:
        print("ERROR: Empty Matrix. Check that the file exists and all "
              "parameters match (width, block diagonal).", file=sys.stderr)
        return

    arrow = ArrowDecompositionMPI.initialize(comm, n_blocks, to_prev, to_next,
                                             width, n_features, device, blocked, slim)

    wb_logging.log({"actual_ranks": comm.size})
    rng = np.random.default_rng(42 + comm.rank)
    comm.barrier()

    tic = time.perf_counter()
    arrow.load_data_from_blocks(blocks)
    arrow.zero_rhs(width, n_features)
    comm.barrier()
    wb_logging.log({"init_time": time.perf_counter() - tic})

    eng0 = arrow.engines[0]
    for i in range(iterations):
        # fresh features on matrix 0 each iteration (arrow_bench.py:113-116)
        rows = max(eng0.n_owned, 1) * width
        X_p0 = 2 * rng.random((rows, n_features), dtype=datatype) - 1
        eng0.set_features(X_p0)
        comm.barrier()

        fail = False
        try:
            wb_logging.set_iteration_data({"iteration": i})
            tic = time.perf_counter()
            arrow.step()
            toc = time.perf_counter()
            wb_logging.log({"spmm_time": toc - tic})
            if comm.rank == 0:
                print("RANK", comm.rank, "Iteration", i, " -- ", toc - tic,
                      "s", flush=True)
        except Exception as e:
            print("RANK", comm.rank, "EXCEPTION", e, flush=True)
            fail = True
        # collective failure detection (arrow_bench.py:121-134)
        flag = torch.tensor([1 if fail else 0], dtype=torch.int64)
        comm.allreduce_max_(flag)
        if int(flag.item()):
            print("RANK", comm.rank, "FAILED", flush=True)
            break

    wb_logging.finish()
    comm.barrier()
    return arrow

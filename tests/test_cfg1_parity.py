"""BASELINE.json configs[0]/[1] parity gate: an 8k-row random sparse matrix
(~80k nnz, seed 42) decomposed at width=2000, features=16 — the GPU engine
against the scipy `--device cpu` path (the designated parity reference,
north_star: <=1e-5 relative fp32)."""
import os
import tempfile

import numpy as np
import pytest

from arrow_matrix_amd import graphio
from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
from arrow_matrix_amd.common import utils
from arrow_matrix_amd.decomposition import arrow_decomposition


def _make_cfg1(tmpdir):
    rng = np.random.default_rng(42)
    A = utils.generate_sparse_matrix(8000, 8000, 80000, np.float32, rng)
    A = A.maximum(A.T).tocsr()  # symmetric adjacency-like
    decomp = arrow_decomposition(A, arrow_width=2000, max_number_of_levels=3,
                                 block_diagonal=True,
                                 rng=np.random.default_rng(0))
    pairs = [(p.graph, p.permutation) for p in decomp]
    prefix = os.path.join(tmpdir, 'cfg1')
    graphio.save_decomposition_new(pairs, prefix, 2000)
    return A, pairs, prefix


def _run(prefix, device, X_engine, k=16, width=2000):
    blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
        None, prefix, width)
    arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width, k,
                                             device=device)
    arrow.load_data_from_blocks(blocks)
    arrow.zero_rhs(width, k)
    arrow.B.set_features(X_engine.copy())
    arrow.step()
    return arrow.B.allgather_result(), nb


def test_cfg1_cpu_matches_golden():
    with tempfile.TemporaryDirectory() as td:
        A, pairs, prefix = _make_cfg1(td)
        rng = np.random.default_rng(7)
        n = A.shape[0]
        X = (2 * rng.random((n, 16), dtype=np.float32) - 1)
        perm0 = pairs[0][1]
        C, nb = _run(prefix, 'cpu', X[perm0])
        golden = (A @ X)[perm0]
        m = min(C.shape[0], n)
        np.testing.assert_allclose(C[:m], golden[:m], rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
def test_cfg1_gpu_matches_cpu_device_at_1e5():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    with tempfile.TemporaryDirectory() as td:
        A, pairs, prefix = _make_cfg1(td)
        rng = np.random.default_rng(7)
        n = A.shape[0]
        X = (2 * rng.random((n, 16), dtype=np.float32) - 1)
        perm0 = pairs[0][1]
        C_cpu, _ = _run(prefix, 'cpu', X[perm0])
        C_gpu, _ = _run(prefix, 'gpu', X[perm0])
        # north_star acceptance: <=1e-5 relative fp32 vs the cpu path
        scale = np.abs(C_cpu).max()
        np.testing.assert_allclose(C_gpu, C_cpu, rtol=1e-5, atol=1e-5 * scale)

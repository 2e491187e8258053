"""ArrowMPI banded mode (`--slim False --blocked False`): parity against the
golden compute_spmm == A @ X (the reference's own gate,
test_arrowmpi.py:342-398) on banded synthetic decompositions."""
import os
import tempfile

import numpy as np
import pytest

from arrow_matrix_amd import graphio, synth
from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
from oracle import compute_spmm, slim_arrow_spmm


def _run_banded(n_blocks, width, k, seed, device='cpu', iters=1):
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5,
                                             seed=seed, block_diagonal=False)
    n = n_blocks[0] * width
    rng = np.random.default_rng(300 + seed)
    X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)

    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, width, block_diagonal=False)
        blocks, nb, to_prev, to_next = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width, is_block_diagonal=False)
        np.testing.assert_array_equal(nb, n_blocks)
        arrow = ArrowDecompositionMPI.initialize(None, nb, to_prev, to_next,
                                                 width, k, device=device,
                                                 block_diagonal=False, slim=False)
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)
        perm0 = decomp[0][1]
        arrow.B.set_features(X_orig[perm0].copy())
        results, goldens = [], []
        golden_X = X_orig
        for _ in range(iters):
            arrow.step()
            results.append(arrow.B.allgather_result().copy())
            goldens.append(compute_spmm(decomp, golden_X)[perm0])
            golden_X = compute_spmm(decomp, golden_X)
            arrow.B.set_features(arrow.B.result_tile())
    return results, goldens


@pytest.mark.parametrize("n_blocks,width,k,seed", [
    ([3], 6, 4, 0), ([5], 5, 8, 1), ([4, 2], 5, 4, 2), ([1], 4, 3, 3),
])
def test_banded_cpu(n_blocks, width, k, seed):
    results, goldens = _run_banded(n_blocks, width, k, seed)
    np.testing.assert_allclose(results[0], goldens[0], rtol=1e-4, atol=1e-4)


def test_banded_oracle_single_matrix():
    """slim_arrow_spmm with halo blocks == B @ X."""
    width, nb, k = 6, 4, 5
    decomp = synth.synth_arrow_decomposition(width, [nb], avg_deg=5, seed=4,
                                             block_diagonal=False)
    B, _ = decomp[0]
    from arrow_matrix_amd.graphio import split_matrix_to_blocks
    blocks = split_matrix_to_blocks(B, width)
    rng = np.random.default_rng(0)
    X = (2 * rng.random((nb * width, k)) - 1).astype(np.float32)
    tiles = [X[r * width:(r + 1) * width] for r in range(nb)]
    C = np.concatenate(slim_arrow_spmm(blocks, tiles))
    np.testing.assert_allclose(C, B @ X, rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
def test_banded_gpu():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    results, goldens = _run_banded([4], 64, 16, 5, device='gpu', iters=2)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=2e-4, atol=2e-4)


@pytest.mark.gpu
def test_banded_gpu_multi_part():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    results, goldens = _run_banded([3, 2], 40, 8, 6, device='gpu')
    np.testing.assert_allclose(results[0], goldens[0], rtol=2e-4, atol=2e-4)

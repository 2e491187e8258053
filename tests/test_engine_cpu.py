"""End-to-end engine parity (single process, cpu device = the reference's
scipy path) against the oracle golden compute_spmm == A @ X."""
import os
import tempfile

import numpy as np
import pytest
import torch

from arrow_matrix_amd import graphio, synth
from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
from oracle import compute_spmm


def _run_engine(decomp, width, n_blocks, k, iters=1, device='cpu', seed=0):
    """Save -> load -> initialize -> iterate; returns list of gathered C per
    iteration plus the goldens (in part-0 ordering)."""
    n = n_blocks[0] * width
    rng = np.random.default_rng(seed)
    X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)

    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, width)
        blocks, nb, to_prev, to_next = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width, is_block_diagonal=True)
        np.testing.assert_array_equal(nb, n_blocks)
        arrow = ArrowDecompositionMPI.initialize(None, nb, to_prev, to_next,
                                                 width, k, device=device)
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)

        perm0 = np.argsort(np.argsort(decomp[0][1]))  # identity-safe; see below
        # engine X is in part-0 order
        perms_padded = decomp[0][1]
        X_engine = X_orig[perms_padded]
        arrow.B.set_features(X_engine.copy())

        results, goldens = [], []
        golden_X = X_orig
        for _ in range(iters):
            arrow.step()
            C = arrow.B.allgather_result()
            results.append(C.copy())
            golden_C = compute_spmm(decomp, golden_X)[perms_padded]
            goldens.append(golden_C)
            # next iteration: X := C on the engine, golden_X := A @ golden_X
            arrow.B.set_features(arrow.B.result_tile())
            golden_X = compute_spmm(decomp, golden_X)
        return results, goldens


@pytest.mark.parametrize("n_blocks,width,k,seed", [
    ([3], 5, 4, 0),
    ([4], 6, 16, 1),
    ([1], 4, 3, 2),
    ([4, 2], 5, 8, 3),
    ([3, 3], 4, 5, 4),
    ([4, 3, 2], 4, 6, 5),
])
def test_engine_single_process_cpu(n_blocks, width, k, seed):
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=seed)
    results, goldens = _run_engine(decomp, width, n_blocks, k, iters=1, seed=seed)
    np.testing.assert_allclose(results[0], goldens[0], rtol=2e-5, atol=1e-5)


def test_engine_iterated_cpu():
    """3 chained iterations (the reference's test_decomposition chains 3,
    test_arrowmpi.py:164-166)."""
    n_blocks, width, k = [3, 2], 5, 4
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=4, seed=7)
    results, goldens = _run_engine(decomp, width, n_blocks, k, iters=3, seed=7)
    for C, G in zip(results, goldens):
        np.testing.assert_allclose(C, G, rtol=1e-4, atol=1e-4)


def test_engine_with_hub_rows():
    """Power-law first block-row (hub vertices) — exercises the long-row
    path of the kernel on GPU; here checks cpu parity."""
    n_blocks, width, k = [4], 8, 4
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=3, seed=11,
                                             hub_rows=2, hub_deg=20)
    results, goldens = _run_engine(decomp, width, n_blocks, k, seed=11)
    np.testing.assert_allclose(results[0], goldens[0], rtol=2e-5, atol=1e-5)


def test_gpu_device_without_gpu_fails_loudly():
    """The product GPU path must never fall back to CPU."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from arrow_matrix_amd.backends import make_backend
    from arrow_matrix_amd.hip import ArrowSpmmError
    with pytest.raises((ArrowSpmmError, RuntimeError)):
        make_backend('gpu')


def test_engine_cpu_float64():
    """float64 on the cpu device (the reference's -t float64 option,
    spmm_petsc_main.py / datatype param): full-precision scipy path."""
    from arrow_matrix_amd import graphio
    n_blocks, width, k = [3], 6, 4
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=77)
    n = n_blocks[0] * width
    rng = np.random.default_rng(77)
    X = (2 * rng.random((n, k)) - 1)  # float64
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, width)
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width, datatype=np.float64)
        arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width, k,
                                                 device='cpu')
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k, dtype=np.float64)
        perm0 = decomp[0][1]
        arrow.B.set_features(X[perm0].copy())
        arrow.step()
        C = arrow.B.allgather_result()
        assert C.dtype == np.float64
        A = synth.recompose(decomp).astype(np.float64)
        golden = (A @ X)[perm0]
        np.testing.assert_allclose(C, golden, rtol=1e-12, atol=1e-12)


def test_petsc_cpu_float64():
    from scipy import sparse
    from arrow_matrix_amd.matrix_slice import MatrixSlice
    from arrow_matrix_amd.spmm_petsc import SpmmPETSc
    rng = np.random.RandomState(3)
    A = sparse.csr_matrix(sparse.random(20, 20, density=0.3, random_state=rng,
                                        format='csr'), dtype=np.float64)
    ms = MatrixSlice.initialize(None, A)
    eng = SpmmPETSc(None, ms, device='cpu', dtype=np.float64)
    X = rng.rand(20, 3)
    Y = eng.spmm(X)
    assert Y.dtype == torch.float64
    np.testing.assert_allclose(Y.numpy(), A @ X, rtol=1e-12, atol=1e-12)


def test_engine_folded_permutation_cpu(monkeypatch):
    """Folded-permutation mode (ARROW_FOLD=1, single process, L>1): parts
    i>=1 re-indexed into part 0's numbering, no per-step exchange — results
    must match the sequential cascade / golden (arrow_dec.py::_build_folded)."""
    monkeypatch.setenv('ARROW_FOLD', '1')
    for n_blocks, seed in ([4, 2], 3), ([3, 3], 4), ([4, 3, 2], 5):
        decomp = synth.synth_arrow_decomposition(5, n_blocks, avg_deg=5, seed=seed)
        results, goldens = _run_engine(decomp, 5, n_blocks, 8, iters=3, seed=seed)
        for C, G in zip(results, goldens):
            np.testing.assert_allclose(C, G, rtol=1e-4, atol=1e-4)


def test_engine_folded_engages_and_matches_sequential(monkeypatch):
    """The fold path must actually engage (structures built, exchange
    skipped) and agree with the sequential path on the same input."""
    import tempfile
    width, n_blocks, k = 6, [4, 2], 5
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=21)
    n = n_blocks[0] * width
    rng = np.random.default_rng(21)
    X = (2 * rng.random((n, k)) - 1).astype(np.float32)

    def run(fold):
        monkeypatch.setenv('ARROW_FOLD', '1' if fold else '0')
        with tempfile.TemporaryDirectory() as td:
            prefix = os.path.join(td, 'g')
            graphio.save_decomposition_new(decomp, prefix, width)
            blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
                None, prefix, width)
            arrow = ArrowDecompositionMPI.initialize(None, nb, tp, tn, width,
                                                     k, device='cpu')
            arrow.load_data_from_blocks(blocks)
            arrow.zero_rhs(width, k)
            assert (arrow._folded is not None) == fold
            arrow.B.set_features(X[decomp[0][1]].copy())
            outs = []
            for _ in range(2):
                arrow.step()
                outs.append(arrow.B.allgather_result().copy())
                arrow.B.set_features(arrow.B.result_tile())
            return outs

    folded, seq = run(True), run(False)
    for F, S in zip(folded, seq):
        np.testing.assert_allclose(F, S, rtol=2e-6, atol=2e-6)


def test_engine_row_folded_cpu(monkeypatch):
    """ROW-fold mode (ARROW_FOLD=2): forward exchange materialised, only
    the backward cascade folded into part-i launches — must match golden."""
    monkeypatch.setenv('ARROW_FOLD', '2')
    for n_blocks, seed in ([4, 2], 3), ([3, 3], 4), ([4, 3, 2], 5):
        decomp = synth.synth_arrow_decomposition(5, n_blocks, avg_deg=5, seed=seed)
        results, goldens = _run_engine(decomp, 5, n_blocks, 8, iters=3, seed=seed)
        for C, G in zip(results, goldens):
            np.testing.assert_allclose(C, G, rtol=1e-4, atol=1e-4)

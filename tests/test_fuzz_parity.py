"""Randomised parity sweep: many random decomposition shapes through the
full public path (save -> load -> initialize -> step -> allgather) against
the golden compute_spmm == A @ X, on CPU (the reference-designated parity
path)."""
import os
import tempfile

import numpy as np
import pytest

from arrow_matrix_amd import graphio, synth
from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
from oracle import compute_spmm

CASES = []
_rng = np.random.default_rng(2024)
for i in range(12):
    L = int(_rng.integers(1, 4))
    nb0 = int(_rng.integers(1, 6))
    nbs = [nb0] + sorted(
        (int(_rng.integers(1, nb0 + 1)) for _ in range(L - 1)), reverse=True)
    CASES.append(dict(
        n_blocks=nbs,
        width=int(_rng.integers(3, 12)),
        k=int(_rng.integers(1, 9)),
        seed=1000 + i,
        banded=bool(_rng.integers(0, 2)) and L == 1,  # banded single-part
        one_based=bool(_rng.integers(0, 2)),
        hub_rows=int(_rng.integers(0, 3)),
    ))


@pytest.mark.parametrize("case", CASES)
def test_fuzz_case(case):
    blocked = not case['banded']
    decomp = synth.synth_arrow_decomposition(
        case['width'], case['n_blocks'], avg_deg=int(np.random.default_rng(
            case['seed']).integers(2, 8)), seed=case['seed'],
        block_diagonal=blocked, hub_rows=case['hub_rows'])
    if case['one_based']:
        decomp = [(B, p + 1) for B, p in decomp]
    n = case['n_blocks'][0] * case['width']
    rng = np.random.default_rng(case['seed'] + 1)
    X = (2 * rng.random((n, case['k'])) - 1).astype(np.float32)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, case['width'],
                                       block_diagonal=blocked)
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, case['width'], is_block_diagonal=blocked)
        arrow = ArrowDecompositionMPI.initialize(
            None, nb, tp, tn, case['width'], case['k'], device='cpu',
            block_diagonal=blocked, slim=blocked)
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(case['width'], case['k'])
        perm0 = np.asarray(decomp[0][1], dtype=np.int64)
        if case['one_based']:
            perm0 = perm0 - 1
        arrow.B.set_features(X[perm0].copy())
        arrow.step()
        C = arrow.B.allgather_result()
        zero_decomp = [(B, (p - 1 if case['one_based'] else p))
                       for B, p in decomp]
        golden = compute_spmm(zero_decomp, X)[perm0]
        np.testing.assert_allclose(C, golden, rtol=1e-4, atol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize("case", CASES[:8])
def test_fuzz_case_gpu(case):
    """The same randomised sweep through the GPU engine."""
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    blocked = not case['banded']
    decomp = synth.synth_arrow_decomposition(
        case['width'], case['n_blocks'], avg_deg=int(np.random.default_rng(
            case['seed']).integers(2, 8)), seed=case['seed'],
        block_diagonal=blocked, hub_rows=case['hub_rows'])
    if case['one_based']:
        decomp = [(B, p + 1) for B, p in decomp]
    n = case['n_blocks'][0] * case['width']
    rng = np.random.default_rng(case['seed'] + 1)
    X = (2 * rng.random((n, case['k'])) - 1).astype(np.float32)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(decomp, prefix, case['width'],
                                       block_diagonal=blocked)
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, case['width'], is_block_diagonal=blocked)
        arrow = ArrowDecompositionMPI.initialize(
            None, nb, tp, tn, case['width'], case['k'], device='gpu',
            block_diagonal=blocked, slim=blocked)
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(case['width'], case['k'])
        perm0 = np.asarray(decomp[0][1], dtype=np.int64)
        if case['one_based']:
            perm0 = perm0 - 1
        arrow.B.set_features(X[perm0].copy())
        arrow.step()
        C = arrow.B.allgather_result()
        zero_decomp = [(B, (p - 1 if case['one_based'] else p))
                       for B, p in decomp]
        golden = compute_spmm(zero_decomp, X)[perm0]
        np.testing.assert_allclose(C, golden, rtol=2e-4, atol=2e-4)


@pytest.mark.parametrize("case", [c for c in CASES if len(c['n_blocks']) > 1])
def test_fuzz_case_folded(case, monkeypatch):
    """The same randomised L>1 sweep through the FOLDED permutation path
    (arrow_dec._build_folded) on CPU."""
    monkeypatch.setenv('ARROW_FOLD', '1')
    test_fuzz_case(case)


@pytest.mark.parametrize("case", [c for c in CASES if len(c['n_blocks']) > 1])
def test_fuzz_case_row_folded(case, monkeypatch):
    """The L>1 sweep through the ROW-fold path (backward cascade folded,
    forward exchange materialised) on CPU."""
    monkeypatch.setenv('ARROW_FOLD', '2')
    test_fuzz_case(case)

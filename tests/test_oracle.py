"""Oracle self-consistency: the sequential restatements reproduce
A @ X (scipy) on synthetic arrow decompositions — the reference's own
exactness invariant (test_arrowdecomposition.py:93-100)."""
import numpy as np
import pytest

from arrow_matrix_amd import synth, tables
from oracle import compute_spmm, slim_arrow_spmm, decomposition_step


def _golden(decomp, X):
    A = synth.recompose(decomp)
    return A @ X


@pytest.mark.parametrize("n_blocks,width,seed", [
    ([3], 5, 0), ([4], 8, 1), ([1], 6, 2),
    ([4, 2], 5, 3), ([3, 3], 4, 4), ([4, 3, 2], 4, 5),
])
def test_compute_spmm_equals_recomposed_product(n_blocks, width, seed):
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=seed)
    n = n_blocks[0] * width
    rng = np.random.default_rng(100 + seed)
    X = (2 * rng.random((n, 8)) - 1).astype(np.float32)
    C = compute_spmm(decomp, X)
    np.testing.assert_allclose(C, _golden(decomp, X), rtol=2e-5, atol=1e-5)


@pytest.mark.parametrize("nb,width,k,seed", [(1, 4, 3, 0), (3, 5, 7, 1), (5, 6, 16, 2)])
def test_slim_arrow_spmm_single_matrix(nb, width, k, seed):
    """Single-part slim dataflow == B @ X."""
    decomp = synth.synth_arrow_decomposition(width, [nb], avg_deg=5, seed=seed)
    B, perm = decomp[0]
    from arrow_matrix_amd.graphio import split_matrix_to_blocks
    blocks = split_matrix_to_blocks(B, width)
    rng = np.random.default_rng(seed)
    X = (2 * rng.random((nb * width, k)) - 1).astype(np.float32)
    tiles = [X[r * width:(r + 1) * width] for r in range(nb)]
    out = slim_arrow_spmm(blocks, tiles)
    C = np.concatenate(out)
    np.testing.assert_allclose(C, B @ X, rtol=2e-5, atol=1e-5)


@pytest.mark.parametrize("n_blocks,width,k,seed", [
    ([3, 2], 4, 5, 0), ([4, 4], 5, 3, 1), ([4, 3, 2], 4, 6, 2), ([2], 5, 4, 3),
])
def test_decomposition_step_equals_compute_spmm(n_blocks, width, k, seed):
    """Full sequential step (forward + spmm + backward) == compute_spmm, in
    part-0 ordering — the reference's distributed-parity golden
    (test_arrowmpi.py:290)."""
    decomp = synth.synth_arrow_decomposition(width, n_blocks, avg_deg=5, seed=seed)
    n = n_blocks[0] * width
    rng = np.random.default_rng(200 + seed)
    X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)

    perms = [p for _, p in decomp]
    perms_p, to_prev, to_next = tables.pad_and_compose_permutations(
        perms, np.asarray(n_blocks), width)

    from arrow_matrix_amd.graphio import split_matrix_to_blocks
    parts_blocks = []
    for i, (B, _) in enumerate(decomp):
        g = split_matrix_to_blocks(B, width)
        g = [row[:n_blocks[i]] for row in g[:n_blocks[i]]]
        parts_blocks.append(g)

    # X in part-0 order, distributed as per-rank tiles; other parts zero
    X0 = X_orig[perms_p[0]]
    X_tiles = [[X0[r * width:(r + 1) * width].copy() for r in range(n_blocks[0])]]
    C_tiles = [[np.zeros((width, k), np.float32) for _ in range(n_blocks[0])]]
    for i in range(1, len(n_blocks)):
        X_tiles.append([np.zeros((width, k), np.float32) for _ in range(n_blocks[i])])
        C_tiles.append([np.zeros((width, k), np.float32) for _ in range(n_blocks[i])])

    decomposition_step(parts_blocks, n_blocks, to_prev, to_next, width,
                       X_tiles, C_tiles)

    C = np.concatenate(C_tiles[0])
    golden = compute_spmm(decomp, X_orig)[perms_p[0]]
    np.testing.assert_allclose(C, golden, rtol=2e-5, atol=1e-5)

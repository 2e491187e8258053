"""Producer invariants — the reference's own decomposition tests restated
(test_arrowdecomposition.py): permutation validity (:44-48), edge-disjoint
exact reconstruction (:93-99), band/block criterion (:69-77), SpMM parity
via compute_spmm (:139-156). The linearisation quality itself is unpinned
(randomised)."""
import numpy as np
import pytest
from scipy import sparse

from arrow_matrix_amd.decomposition import arrow_decomposition, get_arrow_width
from oracle import compute_spmm


def _random_graph(n, avg_deg, seed, power_law=False):
    rng = np.random.default_rng(seed)
    m = n * avg_deg // 2
    if power_law:
        u = rng.random(m)
        src = (u * u * n).astype(np.int64).clip(0, n - 1)
    else:
        src = rng.integers(0, n, m)
    dst = rng.integers(0, n, m)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    A = sparse.csr_matrix((np.ones(src.size, np.float32), (src, dst)), shape=(n, n))
    A = sparse.csr_matrix(A.maximum(A.T))
    A.data[:] = 1.0
    return A


@pytest.mark.parametrize("n,deg,width_c,block_diagonal,levels,seed", [
    (64, 4, 4, True, 100, 0),
    (128, 8, 8, True, 100, 1),
    (128, 6, 10, False, 100, 2),   # banded mode
    (200, 5, 6, True, 2, 3),       # level cap -> best-effort last part
    (256, 6, 8, True, 100, 4),
])
def test_decomposition_invariants(n, deg, width_c, block_diagonal, levels, seed):
    A = _random_graph(n, deg, seed, power_law=True)
    width = n // width_c + 1
    rng = np.random.default_rng(seed)
    decomp = arrow_decomposition(A, width, max_number_of_levels=levels,
                                 block_diagonal=block_diagonal, rng=rng)
    assert len(decomp) >= 1

    # permutations are permutations (test_arrowdecomposition.py:44-48)
    for part in decomp:
        assert np.array_equal(np.sort(part.permutation), np.arange(n))

    # exact reconstruction: A == sum P_i^T B_i P_i (:93-99)
    val_A = None
    for part in decomp:
        inv = np.argsort(part.permutation)
        P = sparse.csr_matrix((np.ones(n, np.float32), inv, np.arange(n + 1)),
                              shape=(n, n))
        term = P @ part.graph @ P.T
        val_A = term if val_A is None else val_A + term
    diff = sparse.csr_matrix(val_A) - A
    assert diff.nnz == 0 or abs(diff).max() < 1e-6

    # edge-disjointness: total nnz preserved (:64-67; exact since every edge
    # lands in exactly one part and reconstruction is exact)
    assert sum(p.graph.nnz for p in decomp) == A.nnz

    # band/block criterion for every non-final part (:69-77)
    for part in decomp[:-1]:
        coo = part.graph.tocoo()
        w = part.arrow_width
        if block_diagonal:
            ok = ((coo.row // w) == (coo.col // w)) | (coo.row < w) | (coo.col < w)
        else:
            ok = (np.abs(coo.row - coo.col) <= w) | (coo.row < w) | (coo.col < w)
        assert ok.all()
    # last part: reported actual width covers everything (:57-62)
    last = decomp[-1]
    assert get_arrow_width(last.graph, last.arrow_width) == last.arrow_width

    # SpMM parity: compute_spmm(decomp, X) == A @ X (:139-156)
    X = (2 * rng.random((n, 8)) - 1).astype(np.float32)
    C = compute_spmm([(p.graph, p.permutation) for p in decomp], X)
    np.testing.assert_allclose(C, A @ X, rtol=2e-5, atol=1e-5)


def test_head_is_highest_degree():
    """The arrow head holds the width highest-degree vertices
    (decomposition.py:253-262)."""
    A = _random_graph(150, 6, 9, power_law=True)
    width = 20
    decomp = arrow_decomposition(A, width, rng=np.random.default_rng(0))
    head = decomp[0].permutation[:width]
    deg = A.getnnz(1)
    assert deg[head].min() >= np.sort(deg)[::-1][width - 1] - 1e-9


def test_end_to_end_decompose_then_engine():
    """Producer -> on-disk files -> loader -> engine step == A @ X."""
    import os
    import tempfile
    from arrow_matrix_amd import graphio
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI

    n, width = 120, 24
    A = _random_graph(n, 5, 12, power_law=True)
    decomp = arrow_decomposition(A, width, max_number_of_levels=5,
                                 block_diagonal=True,
                                 rng=np.random.default_rng(1))
    pairs = [(p.graph, p.permutation) for p in decomp]
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(pairs, prefix, width)
        blocks, nb, to_prev, to_next = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width)
        k = 6
        arrow = ArrowDecompositionMPI.initialize(None, nb, to_prev, to_next,
                                                 width, k, device='cpu')
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)
        rng = np.random.default_rng(2)
        n_pad = int(nb[0]) * width
        X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)
        perm0 = np.asarray(decomp[0].permutation)
        # engine X in part-0 order, padded region zero
        X_eng = np.zeros((n_pad, k), np.float32)
        X_eng[:min(n_pad, n)] = X_orig[perm0][:min(n_pad, n)]
        arrow.B.set_features(X_eng.copy())
        arrow.step()
        C = arrow.B.allgather_result()
        golden = (A @ X_orig)[perm0]
        np.testing.assert_allclose(C[:min(n_pad, n)], golden[:min(n_pad, n)],
                                   rtol=1e-4, atol=1e-4)


def test_end_to_end_ragged_tail():
    """n not a multiple of width: the loader's edge-padded square last
    block (graphio.py:394-399) through producer -> files -> engine."""
    import os
    import tempfile
    from arrow_matrix_amd import graphio
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI

    n, width = 130, 24   # ceil(130/24) = 6 blocks, last block 10 rows ragged
    A = _random_graph(n, 6, 21, power_law=True)
    decomp = arrow_decomposition(A, width, max_number_of_levels=4,
                                 block_diagonal=True,
                                 rng=np.random.default_rng(4))
    pairs = [(p.graph, p.permutation) for p in decomp]
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'g')
        graphio.save_decomposition_new(pairs, prefix, width)
        blocks, nb, to_prev, to_next = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, width)
        k = 5
        arrow = ArrowDecompositionMPI.initialize(None, nb, to_prev, to_next,
                                                 width, k, device='cpu')
        arrow.load_data_from_blocks(blocks)
        arrow.zero_rhs(width, k)
        rng = np.random.default_rng(5)
        n_pad = int(nb[0]) * width
        X_orig = (2 * rng.random((n, k)) - 1).astype(np.float32)
        perm0 = np.asarray(decomp[0].permutation)
        X_eng = np.zeros((n_pad, k), np.float32)
        m = min(n_pad, n)
        X_eng[:m] = X_orig[perm0][:m]
        arrow.B.set_features(X_eng.copy())
        arrow.step()
        C = arrow.B.allgather_result()
        golden = (A @ X_orig)[perm0]
        np.testing.assert_allclose(C[:m], golden[:m], rtol=1e-4, atol=1e-4)

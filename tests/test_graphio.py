"""On-disk format + block splitting parity with the reference goldens."""
import os
import tempfile

import numpy as np
import pytest
from scipy import sparse

from arrow_matrix_amd import graphio, synth

GOLDEN = os.path.join(os.path.dirname(__file__), 'golden', 'reference_katsets.npz')


@pytest.fixture(scope='module')
def golden():
    return np.load(GOLDEN)


def test_format_path_naming():
    # naming contract graphio.py:38-70
    assert graphio.format_path('p', 5, 2, True, graphio.DecompositionFileType.indptr_npy) \
        == 'p_B_5_2_bd_indptr.npy'
    assert graphio.format_path('p', 5, None, False, graphio.DecompositionFileType.data_npy) \
        == 'p_B_5_data.npy'
    assert graphio.format_path('p', 7, 0, True, graphio.DecompositionFileType.permutation_npy) \
        == 'p_B_7_0_bd_permutation.npy'
    assert graphio.format_path('p', 7, 0, True, graphio.DecompositionFileType.nonzero_rows_npy) \
        == 'p_B_7_0_bd_nnzrows.npy'


def test_save_load_roundtrip():
    decomp = synth.synth_arrow_decomposition(5, [3, 2], avg_deg=4, seed=9)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'x')
        graphio.save_decomposition_new(decomp, prefix, 5)
        loaded = graphio.load_decomposition_new(prefix, 5)
        assert len(loaded) == 2
        for (B0, p0), (B1, p1) in zip(decomp, loaded):
            # the loader infers the column count from indices (same quirk as
            # reference graphio.py:302) — compare on the common shape
            a = sparse.csr_matrix(B0)
            b = sparse.csr_matrix((B1.data, B1.indices, B1.indptr), shape=a.shape)
            diff = a - b
            assert diff.nnz == 0 or abs(diff).max() == 0
            np.testing.assert_array_equal(p0, p1)
        # mem_map returns raw arrays
        mm = graphio.load_decomposition_new(prefix, 5, mem_map=True)
        data, indices, indptr = mm[0][0]
        np.testing.assert_array_equal(indptr, sparse.csr_matrix(decomp[0][0]).indptr)


def test_missing_data_file_becomes_ones():
    decomp = synth.synth_arrow_decomposition(4, [2], avg_deg=3, seed=1)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'x')
        graphio.save_decomposition_new(decomp, prefix, 4)
        os.remove(graphio.format_path(prefix, 4, 0, True,
                                      graphio.DecompositionFileType.data_npy))
        loaded = graphio.load_decomposition_new(prefix, 4)
        assert np.all(loaded[0][0].data == 1.0)


def test_split_matrix_to_blocks_vs_golden(golden):
    n = int(golden['split_n_cases'][0])
    for ci in range(n):
        A = sparse.csr_matrix(golden[f'split_{ci}_dense'].astype(np.float32))
        bs = int(golden[f'split_{ci}_bs'][0])
        gi, gj = (int(x) for x in golden[f'split_{ci}_grid'])
        blocks = graphio.split_matrix_to_blocks(A, bs)
        assert len(blocks) == gi and len(blocks[0]) == gj
        for i in range(gi):
            for j in range(gj):
                key = f'split_{ci}_blk_{i}_{j}'
                if key in golden:
                    assert blocks[i][j] is not None
                    assert tuple(blocks[i][j].shape) == tuple(golden[key + '_shape'])
                    np.testing.assert_array_equal(blocks[i][j].toarray(), golden[key])
                else:
                    assert blocks[i][j] is None


def test_legacy_npz_roundtrip_and_cli_path():
    """Legacy .npz on-disk format (reference graphio.py:103-117,194-248)."""
    from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
    decomp = synth.synth_arrow_decomposition(4, [3], avg_deg=4, seed=21)
    with tempfile.TemporaryDirectory() as td:
        prefix = os.path.join(td, 'x')
        graphio.save_decomposition(decomp, prefix, 4)
        loaded = graphio.load_decomposition(prefix, 4)
        assert len(loaded) == 1
        diff = sparse.csr_matrix(decomp[0][0]) - sparse.csr_matrix(loaded[0][0])
        assert diff.nnz == 0 or abs(diff).max() == 0
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, prefix, 4, use_npy=False)
        assert int(nb[0]) == 3

"""Loader parity: my per-rank blocks/permutation slices vs the payloads the
REFERENCE loader scattered (captured send stream in the golden fixtures,
tests/golden/gen_golden.py — protocol: _send_block tags 0-3,
_send_permutation tag 4, arrow_dec_mpi.py:889-902)."""
import os
import tempfile

import numpy as np
import pytest
from scipy import sparse

from arrow_matrix_amd import graphio, synth
from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
from arrow_matrix_amd.comm import Comm

GOLDEN = os.path.join(os.path.dirname(__file__), 'golden', 'reference_katsets.npz')


class _RankComm(Comm):
    def __init__(self, rank, size):
        self.rank = rank
        self.size = size


@pytest.fixture(scope='module')
def golden():
    return np.load(GOLDEN)


def _decode_sends(golden, ci):
    """Group the captured send stream by destination, in order; decode
    blocks (tag 0-3 quadruples) and permutations (tag 4)."""
    n = int(golden[f'load_{ci}_n_sends'][0])
    by_dest = {}
    for si in range(n):
        dest, tag = (int(x) for x in golden[f'load_{ci}_send_{si}_hdr'])
        by_dest.setdefault(dest, []).append((tag, golden[f'load_{ci}_send_{si}_buf']))
    decoded = {}
    for dest, msgs in by_dest.items():
        blocks, perms = [], []
        i = 0
        while i < len(msgs):
            tag, buf = msgs[i]
            if tag == 0:
                rows, nnz = int(buf[0]), int(buf[1])
                assert msgs[i + 1][0] == 1 and msgs[i + 2][0] == 2 and msgs[i + 3][0] == 3
                data, indices, indptr = msgs[i + 1][1], msgs[i + 2][1], msgs[i + 3][1]
                b = sparse.csr_matrix((data, indices.astype(np.int32),
                                       indptr.astype(np.int64)), shape=(rows, rows))
                b.sum_duplicates()
                b.sort_indices()
                blocks.append(b)
                i += 4
            elif tag == 4:
                perms.append(buf.astype(np.int64))
                i += 1
            else:
                raise AssertionError(f"unexpected tag {tag}")
        decoded[dest] = (blocks, perms)
    return decoded


def _assert_blocks_equal(a, b, msg=''):
    assert a is not None and b is not None, msg
    assert a.shape == b.shape, f"{msg}: {a.shape} vs {b.shape}"
    d = sparse.csr_matrix(a) - sparse.csr_matrix(b)
    assert abs(d).max() if d.nnz else 0 == 0, msg


def test_loader_matches_reference_scatter(golden):
    n_cases = int(golden['load_n_cases'][0])
    for ci in range(n_cases):
        meta = [int(x) for x in golden[f'load_{ci}_meta']]
        width, L, one_based, seed = meta[:4]
        banded = bool(meta[4]) if len(meta) > 4 else False
        n_blocks_ref = golden[f'load_{ci}_n_blocks']
        # regenerate the exact same decomposition + files
        nb_list = list(n_blocks_ref)
        decomp = synth.synth_arrow_decomposition(width, nb_list, avg_deg=4,
                                                 seed=seed,
                                                 block_diagonal=not banded)
        if one_based:
            decomp = [(B, p + 1) for B, p in decomp]
        with tempfile.TemporaryDirectory() as td:
            prefix = os.path.join(td, 'g')
            graphio.save_decomposition_new(decomp, prefix, width,
                                           block_diagonal=not banded)
            # my shared-rank loader: P ranks = max blocks so that each rank
            # owns exactly one block-row of every part
            P = int(max(nb_list))
            per_rank = []
            for r in range(P):
                blocks, n_blocks, to_prev, to_next = \
                    ArrowDecompositionMPI.load_decomposition_new(
                        _RankComm(r, P), prefix, width,
                        is_block_diagonal=not banded)
                np.testing.assert_array_equal(n_blocks, n_blocks_ref)
                per_rank.append((blocks, to_prev, to_next))

        decoded = _decode_sends(golden, ci)

        # rank 0 of the reference keeps A_00 of part 0 + its to_next slice
        my_blocks0, my_tp0, my_tn0 = per_rank[0]
        ref_blk00 = golden[f'load_{ci}_rank0_blk_0_0']
        _assert_blocks_equal(my_blocks0[0][0][0].toarray(), ref_blk00, 'rank0 A_00')
        if f'load_{ci}_rank0_to_next' in golden:
            np.testing.assert_array_equal(my_tn0[0], golden[f'load_{ci}_rank0_to_next'])

        base = 0
        for i, nb in enumerate(nb_list):
            # reference rank layout: slim -> nb ranks per part (dest base+r);
            # non-slim -> 2nb-1 ranks (row-0 columns at base+c, block-row r
            # at base+nb-1+r, arrow_dec_mpi.py:753-823)
            for r in range(nb):
                dest = base + r if not banded else (base + nb - 1 + r if r > 0
                                                    else base)
                if dest == 0:
                    continue
                blocks_d, perms_d = decoded[dest]
                my_blocks, my_tp, my_tn = per_rank[r]
                grid = my_blocks[i]
                if r == 0:
                    # reference base rank of part i>0 receives A_00 (+ perms)
                    exp = [grid[0][0]]
                elif banded:
                    exp = [grid[r][0], grid[r][r]]
                    if r > 1:
                        exp.append(grid[r][r - 1])
                    if r < nb - 1:
                        exp.append(grid[r][r + 1])
                else:
                    exp = [grid[0][r], grid[r][0], grid[r][r]]
                assert len(blocks_d) == len(exp), f"dest {dest}: block count"
                for got, mine in zip(blocks_d, exp):
                    _assert_blocks_equal(got.toarray(), mine.toarray(),
                                         f'part {i} block-row {r}')
                # permutation slices: to_prev (i>0) then to_next (i<L-1)
                expected_perms = []
                if i > 0:
                    expected_perms.append(my_tp[i])
                if i < L - 1:
                    expected_perms.append(my_tn[i])
                assert len(perms_d) == len(expected_perms), f"dest {dest}: perm count"
                for got, mine in zip(perms_d, expected_perms):
                    np.testing.assert_array_equal(got, mine)
            if banded:
                # row-0 column tiles at base+1..base+nb-1 (A_0c only)
                for c in range(1, nb):
                    blocks_d, perms_d = decoded[base + c]
                    grid = per_rank[c][0][i]
                    assert len(blocks_d) == 1, f"col dest {base+c}"
                    _assert_blocks_equal(blocks_d[0].toarray(),
                                         grid[0][c].toarray(),
                                         f'part {i} A_0{c}')
            base += nb if not banded else 2 * nb - 1

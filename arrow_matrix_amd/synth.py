"""Numpy-only synthetic arrow-decomposition generator.

Replaces the reference's igraph-based producer (arrow/decomposition.py +
igraph.Graph.Barabasi in arrow_bench.py:28-41) for tests and benchmarks:
instead of decomposing a graph, it directly constructs decomposition parts
B_i with the arrow sparsity structure plus permutations P_i, and the
"original" matrix is defined as A := sum_i P_i^T B_i P_i — exactly the
invariant the reference's own tests verify (test_arrowdecomposition.py:93-99).

Structure of each part (block grid of `width`-sized blocks, block-diagonal
arrow as consumed by ArrowSlimMPI, arrow_slim_mpi.py:298-326):
  - block-row 0: any column block (A_0i),
  - block-row i>0: block-column 0 (A_i0) and the diagonal block (A_ii);
  - banded mode additionally allows blocks (i, i-1) and (i, i+1)
    (ArrowMPI, arrow_mpi.py).

Parts i>0 may be confined to their first m_i = n_blocks[i]*width positions
(trailing zero rows), exercising the reference loader's zero-block cut
(arrow_dec_mpi.py:612-627) and the permutation overflow sentinel
(arrow_dec_mpi.py:740-749). Support vertex sets are nested across parts
(S_0 ⊇ S_1 ⊇ ...), matching real arrow decompositions (each later part covers
leftover edges) — the property that makes the forward feature propagation
exact.
"""
from typing import List, Optional, Sequence, Tuple

import numpy as np
from scipy import sparse


def synth_arrow_decomposition(width: int,
                              n_blocks: Sequence[int],
                              avg_deg: int = 8,
                              seed: int = 42,
                              block_diagonal: bool = True,
                              hub_rows: int = 0,
                              hub_deg: Optional[int] = None,
                              dtype=np.float32
                              ) -> List[Tuple[sparse.csr_matrix, np.ndarray]]:
    """Generate a decomposition [(B_i, perm_i), ...].

    n_blocks[0] defines n = n_blocks[0]*width (part 0 is uncut);
    n_blocks[i] <= n_blocks[0] confines part i's support to its first
    n_blocks[i]*width positions. `hub_rows`/`hub_deg` add heavy rows to
    block-row 0 of part 0 (power-law skew of A_0i, decomposition.py:258-262).
    """
    assert all(b <= n_blocks[0] for b in n_blocks)
    rng = np.random.default_rng(seed)
    n = int(n_blocks[0]) * width
    L = len(n_blocks)

    # Nested support vertex sets V_i = perm_i[0:m_i] with V_i ⊆ V_{i-1}, and
    # permutations placing V_i in the first m_i positions of part i's order.
    perms = []
    prev_V = None
    for i in range(L):
        m_i = int(n_blocks[i]) * width
        if i == 0:
            perm = rng.permutation(n).astype(np.int64)
        else:
            V = prev_V[rng.permutation(prev_V.size)[:m_i]]
            rest = np.setdiff1d(np.arange(n, dtype=np.int64), V)
            perm = np.concatenate([V, rest[rng.permutation(rest.size)]])
        perms.append(perm)
        prev_V = perm[:m_i]

    decomposition = []
    for i in range(L):
        B = _random_arrow_csr(width, int(n_blocks[i]), n, avg_deg, rng,
                              block_diagonal=block_diagonal,
                              hub_rows=hub_rows if i == 0 else 0,
                              hub_deg=hub_deg, dtype=dtype)
        decomposition.append((B, perms[i]))
    return decomposition


def _random_arrow_csr(width: int, nb: int, n: int, avg_deg: int,
                      rng: np.random.Generator, block_diagonal: bool,
                      hub_rows: int = 0, hub_deg: Optional[int] = None,
                      dtype=np.float32) -> sparse.csr_matrix:
    """Random n x n CSR with arrow support confined to the first nb*width
    rows/cols. Every block-row in range gets at least one nonzero (so the
    loader's zero-cut finds exactly nb blocks)."""
    m = nb * width
    rows_list = []
    cols_list = []

    # per-row degrees (at least 1 in the last block-row to pin nb)
    deg = rng.poisson(avg_deg, size=m).astype(np.int64)
    deg = np.maximum(deg, 1)

    row_ids = np.repeat(np.arange(m), deg)
    nnz = row_ids.size
    cols = np.empty(nnz, dtype=np.int64)

    br = row_ids // width  # block-row of each entry
    first = br == 0
    # block-row 0: any column < m
    cols[first] = rng.integers(0, m, size=int(first.sum()))
    # block-row r>0: choose target block among {0, r} (+ {r-1, r+1} banded)
    rest = ~first
    rrest = br[rest]
    if block_diagonal:
        choice = rng.integers(0, 2, size=int(rest.sum()))
        tgt_block = np.where(choice == 0, 0, rrest)
    else:
        choice = rng.integers(0, 4, size=int(rest.sum()))
        tgt_block = np.select(
            [choice == 0, choice == 1, choice == 2, choice == 3],
            [np.zeros_like(rrest), rrest, np.maximum(rrest - 1, 0),
             np.minimum(rrest + 1, nb - 1)])
    cols[rest] = tgt_block * width + rng.integers(0, width, size=int(rest.sum()))

    rows_list.append(row_ids)
    cols_list.append(cols)

    if hub_rows > 0:
        hd = hub_deg or max(avg_deg * 32, m // 4)
        hr = rng.integers(0, width, size=hub_rows)
        for r in hr:
            hcols = rng.integers(0, m, size=hd)
            rows_list.append(np.full(hd, r, dtype=np.int64))
            cols_list.append(hcols)

    rows = np.concatenate(rows_list)
    cols = np.concatenate(cols_list)
    data = (2 * rng.random(rows.size) - 1).astype(dtype)
    B = sparse.csr_matrix((data, (rows, cols)), shape=(n, n), dtype=dtype)
    B.sum_duplicates()
    B.sort_indices()
    return B


def recompose(decomposition) -> sparse.csr_matrix:
    """A = sum_i P_i^T B_i P_i with P_i = I[argsort(perm_i)] — the exactness
    invariant of test_arrowdecomposition.py:93-99 / 127-137."""
    A = None
    for B, perm in decomposition:
        inv = np.argsort(perm)
        n = B.shape[0]
        P = sparse.csr_matrix((np.ones(n, dtype=B.dtype), inv, np.arange(n + 1)),
                              shape=(n, n))
        term = P @ B @ P.T
        A = term if A is None else A + term
    A = sparse.csr_matrix(A)
    A.sum_duplicates()
    A.sort_indices()
    return A

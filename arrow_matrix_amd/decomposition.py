"""arrow_decompose producer — numpy/scipy-native restatement of the
reference's igraph algorithm (arrow/decomposition.py):

  _arrow_linear_order (:253-281): degree-sort; the arrow_width highest-degree
  vertices form the arrow head (first positions), zero-degree singletons go
  last, and the middle is linearised per connected component of a RANDOM
  spanning forest by a DFS that visits small subtrees first
  (linearize_with_random_forest :165-205, _linearize_tree_stack :230-241).

  _arrow_decomposition (:65-144): keep the edges inside the band (|pos_u -
  pos_v| <= width) or block (same width-block) or touching the head
  (pos < width); recurse on the leftover edges up to max_level; the last
  level takes everything and reports the actual width (get_arrow_width
  :57-62).

Output: list of (B_i csr, permutation_i) in the same convention as the
reference (permutation[i] = original id at position i; B_i = adjacency
permuted by argsort(permutation)), saveable by graphio.save_decomposition_new
and consumable by the engine. The linearisation QUALITY is unpinned (the
reference's randomised forest differs run to run); the structural invariants
(exact reconstruction, edge-disjointness, band/block criterion — the
reference's own tests, test_arrowdecomposition.py:44-100) are tested.
"""
from typing import List, Optional, Tuple

import numpy as np
from scipy import sparse
from scipy.sparse import csgraph


class ArrowGraph:
    """Mirror of the reference's ArrowGraph container (decomposition.py:6-29),
    with a scipy CSR in place of the igraph graph."""

    def __init__(self, matrix: sparse.csr_matrix, permutation: np.ndarray,
                 arrow_width: int):
        self.graph = matrix
        self.permutation = np.asarray(permutation, dtype=np.int64)
        # NOTE: despite the name, the reference's same-named attribute counts
        # ZERO-degree vertices (decomposition.py:20: `len([d for d in
        # graph.degree() if d == 0])`) and persists that to _nnzrows.npy
        # (graphio.py:189-191); kept bit-compatible here. degree = row+col
        # nonzeros (igraph counts in+out for directed graphs).
        deg = matrix.getnnz(1) + matrix.getnnz(0)
        self.nonzero_rows = int(np.count_nonzero(deg == 0))
        self.arrow_width = int(arrow_width)

    def __getitem__(self, item):
        if item == 0:
            return self.graph
        if item == 1:
            return self.permutation
        raise IndexError()


def arrow_decomposition(A: sparse.spmatrix, arrow_width: int = 512,
                        max_number_of_levels: int = 2,
                        block_diagonal: bool = False, prune: bool = True,
                        rng: Optional[np.random.Generator] = None
                        ) -> List[ArrowGraph]:
    """Decompose the (square) sparse adjacency A into arrow parts such that
    A = sum_i P_i^T B_i P_i exactly (each edge/nonzero lands in exactly one
    part). Reference arrow_decomposition (decomposition.py:32-54)."""
    A = sparse.csr_matrix(A)
    assert A.shape[0] == A.shape[1]
    assert arrow_width <= A.shape[0]
    rng = rng if rng is not None else np.random.default_rng()

    decomposition: List[ArrowGraph] = []
    _arrow_decomposition(A, arrow_width, decomposition, max_number_of_levels,
                         block_diagonal, prune, rng)
    return decomposition


def get_arrow_width(B: sparse.csr_matrix, initial_width: int) -> int:
    """Smallest width covering all edges outside the head
    (decomposition.py:57-62)."""
    coo = B.tocoo()
    width = initial_width
    mask = (coo.row > width) & (coo.col > width)
    if mask.any():
        width = max(width, int(np.abs(coo.row[mask] - coo.col[mask]).max()))
    return width


def _arrow_decomposition(A, arrow_width, decomposition, max_level,
                         block_diagonal, prune, rng) -> None:
    n = A.shape[0]
    order = _arrow_linear_order(A, arrow_width, rng,
                                deterministic=len(decomposition) + 1 >= max_level)
    inverse = np.argsort(order)

    coo = A.tocoo()
    pu = inverse[coo.row]
    pv = inverse[coo.col]

    if len(decomposition) + 1 < max_level:
        if not block_diagonal:
            in_l1 = np.abs(pu - pv) <= arrow_width  # BAND criterion (:88)
        else:
            in_l1 = (pu // arrow_width) == (pv // arrow_width)  # BLOCK (:95)
        if prune:
            in_l1 |= (pu < arrow_width) | (pv < arrow_width)
        if not in_l1.any():
            in_l1[:] = True
        actual_width = arrow_width
    else:
        in_l1 = np.ones(coo.nnz, dtype=bool)
        actual_width = None  # computed below on the permuted part

    B1 = sparse.csr_matrix(
        (coo.data[in_l1], (pu[in_l1], pv[in_l1])), shape=(n, n))
    B1.sum_duplicates()
    B1.sort_indices()
    if actual_width is None:
        actual_width = get_arrow_width(B1, arrow_width)
    decomposition.append(ArrowGraph(B1, order, actual_width))

    rest = ~in_l1
    if rest.any():
        A2 = sparse.csr_matrix(
            (coo.data[rest], (coo.row[rest], coo.col[rest])), shape=(n, n))
        _arrow_decomposition(A2, arrow_width, decomposition, max_level,
                             block_diagonal, prune, rng)


def _arrow_linear_order(A: sparse.csr_matrix, arrow_width: int,
                        rng: np.random.Generator,
                        deterministic: bool = False) -> np.ndarray:
    """Reference _arrow_linear_order (decomposition.py:253-281)."""
    n = A.shape[0]
    # degree = incident nonzeros (rows; adjacency assumed structurally
    # symmetric, as the reference's undirected graphs are)
    degree = A.getnnz(1)
    by_degree = np.argsort(-degree, kind='stable')
    head = by_degree[:arrow_width]
    tail = by_degree[arrow_width:]
    middle = tail[degree[tail] > 0]
    singletons = tail[degree[tail] == 0]

    order = [head]
    if middle.size:
        sub = A[middle][:, middle]
        sub = sparse.csr_matrix(sub + sub.T)  # symmetrise for the forest
        if deterministic:
            local = _linearize_bfs(sub)
        else:
            local = _linearize_random_forest(sub, rng)
        order.append(middle[local])
    order.append(singletons)
    return np.concatenate(order).astype(np.int64)


def _linearize_bfs(sub: sparse.csr_matrix) -> np.ndarray:
    """Deterministic per-component BFS (linearize_with_ck,
    decomposition.py:147-162)."""
    n = sub.shape[0]
    n_comp, labels = csgraph.connected_components(sub, directed=False)
    out = []
    for c in range(n_comp):
        members = np.flatnonzero(labels == c)
        bfs = csgraph.breadth_first_order(sub, int(members[0]), directed=False,
                                          return_predecessors=False)
        out.append(bfs)
    return np.concatenate(out) if out else np.zeros(0, dtype=np.int64)


def _linearize_random_forest(sub: sparse.csr_matrix,
                             rng: np.random.Generator) -> np.ndarray:
    """Random spanning forest + small-subtree-first DFS
    (linearize_with_random_forest, decomposition.py:165-205)."""
    n = sub.shape[0]
    # random edge weights -> a random spanning forest via MST
    W = sub.tocoo()
    w = rng.random(W.nnz) + 0.1
    weighted = sparse.csr_matrix((w, (W.row, W.col)), shape=sub.shape)
    forest = csgraph.minimum_spanning_tree(weighted)
    forest = forest + forest.T  # undirected adjacency of the forest

    n_comp, labels = csgraph.connected_components(forest + sub * 0,
                                                  directed=False)
    # adjacency lists of the forest
    fcsr = sparse.csr_matrix(forest)
    indptr, indices = fcsr.indptr, fcsr.indices

    order = np.empty(n, dtype=np.int64)
    pos = 0
    visited = np.zeros(n, dtype=bool)
    subtree = np.ones(n, dtype=np.int64)
    parent = np.full(n, -1, dtype=np.int64)

    comp_roots = {}
    for v in range(n):
        c = labels[v]
        if c not in comp_roots:
            comp_roots[c] = v
    for root in comp_roots.values():
        # iterative DFS to establish parents + post-order for subtree sizes
        stack = [root]
        visited[root] = True
        post = []
        while stack:
            v = stack.pop()
            post.append(v)
            for u in indices[indptr[v]:indptr[v + 1]]:
                if not visited[u]:
                    visited[u] = True
                    parent[u] = v
                    stack.append(u)
        for v in reversed(post):
            if parent[v] >= 0:
                subtree[parent[v]] += subtree[v]
        # second DFS emitting the order: children pushed largest-first so the
        # SMALLEST subtree is visited first (_linearize_tree_stack,
        # decomposition.py:230-241)
        stack = [root]
        while stack:
            v = stack.pop()
            order[pos] = v
            pos += 1
            children = [u for u in indices[indptr[v]:indptr[v + 1]]
                        if parent[u] == v]
            children.sort(key=lambda u: subtree[u], reverse=True)
            stack.extend(children)
    assert pos == n
    return order

"""On-disk decomposition format + block splitting.

Re-implements (from scratch, semantics-identical) the reference's
`arrow/common/graphio.py` `.npy`-CSR decomposition format so that decomposed
arrow blocks produced by the reference drop in unchanged:

- file naming:              graphio.py:38-70   (format_path)
- save (new .npy format):   graphio.py:131-191 (save_decomposition_new)
- load (new .npy format):   graphio.py:251-314 (load_decomposition_new;
                            missing *_data.npy -> ones, graphio.py:297-298)
- block splitting:          graphio.py:361-406 (split_matrix_to_blocks:
                            keeps (0, i-1, i, i+1) blocks per row i>0, edge-pads
                            indptr of a short last block-row to block_size and
                            declares it square, graphio.py:389-399)

No igraph dependency: the graph pickle of the reference is not read or
written; only the CSR arrays + permutations (the part of the format the hot
path consumes, arrow_dec_mpi.py:663).
"""
import enum
import os
from typing import List, Optional, Union

import numpy as np
from scipy import sparse


class DecompositionFileType(enum.Enum):
    npz = 1
    indptr_npy = 2
    indices_npy = 3
    data_npy = 4
    permutation_npy = 5
    nonzero_rows_npy = 6


_SUFFIX = {
    DecompositionFileType.npz: ".npz",
    DecompositionFileType.indptr_npy: "_indptr.npy",
    DecompositionFileType.indices_npy: "_indices.npy",
    DecompositionFileType.data_npy: "_data.npy",
    DecompositionFileType.permutation_npy: "_permutation.npy",
    DecompositionFileType.nonzero_rows_npy: "_nnzrows.npy",
}


def format_path(base_path: str, width: int, index: Optional[int], block_diagonal: bool,
                file_type: DecompositionFileType) -> str:
    """Same naming contract as reference graphio.py:38-70."""
    path = f"{base_path}_B_{width}"
    if index is not None:
        path += f"_{index}"
    if block_diagonal:
        path += "_bd"
    return path + _SUFFIX[file_type]


def get_pathname(basename: str, width: int, is_block_diagonal: bool) -> str:
    """Reference graphio.py:498-504."""
    basename = f"{basename}_B"
    if width:
        basename += f"_{width}"
    if is_block_diagonal:
        basename += "_bd"
    return basename


def save_decomposition_new(decomposition, filename: str, width: int,
                           block_diagonal: bool = True) -> None:
    """Save parts as *_indptr/_indices/_data/_permutation .npy files.

    `decomposition` is a list of (csr_matrix, permutation) pairs or
    ArrowGraph-like objects. Each part is saved under its OWN arrow width
    when the entry carries one (`.arrow_width`), exactly as the reference
    does (graphio.py:176-183: the last level can report a larger actual
    width); plain pairs use the caller's `width`. The per-part zero-degree
    counts are persisted to `_nnzrows.npy` under part 0's width
    (reference graphio.py:189-191; the attribute is named `nonzero_rows`
    but counts zero-degree vertices — see decomposition.ArrowGraph).
    File layout identical to reference graphio.py:171-191 (the graph
    pickle / adjacency of the `save_graph` branch is intentionally not
    written; the hot path never reads it).
    """
    widths = []
    zero_deg_counts = []
    for i, entry in enumerate(decomposition):
        B, permutation = entry[0], entry[1]
        w_i = int(getattr(entry, 'arrow_width', width))
        widths.append(w_i)
        B = sparse.csr_matrix(B)
        if hasattr(entry, 'nonzero_rows'):
            zero_deg_counts.append(int(entry.nonzero_rows))
        else:
            deg = B.getnnz(1) + B.getnnz(0)
            zero_deg_counts.append(int(np.count_nonzero(deg == 0)))
        np.save(format_path(filename, w_i, i, block_diagonal, DecompositionFileType.indptr_npy), B.indptr)
        np.save(format_path(filename, w_i, i, block_diagonal, DecompositionFileType.indices_npy), B.indices)
        np.save(format_path(filename, w_i, i, block_diagonal, DecompositionFileType.data_npy), B.data)
        np.save(format_path(filename, w_i, i, block_diagonal, DecompositionFileType.permutation_npy),
                np.asarray(permutation))
    if widths:
        np.save(format_path(filename, widths[0], 0, block_diagonal,
                            DecompositionFileType.nonzero_rows_npy),
                np.asarray(zero_deg_counts, dtype=np.int64))


def load_decomposition_new(filename: str, width: Optional[int] = None,
                           block_diagonal: bool = True, no_permutation: bool = False,
                           mem_map: bool = False):
    """Load decomposition parts; stops at the first missing index.

    Mirrors reference graphio.py:251-314: *_data.npy is optional (-> ones
    float32, graphio.py:297-298). With mem_map=True the CSR is returned as the
    raw (data, indices, indptr) tuple, as the reference does.
    """
    decomposition = []
    i = 0
    while True:
        try:
            f = format_path(filename, width, i, block_diagonal, DecompositionFileType.indptr_npy)
            indptr = np.lib.format.open_memmap(f, mode='r') if mem_map else np.load(f)
            f = format_path(filename, width, i, block_diagonal, DecompositionFileType.indices_npy)
            indices = np.lib.format.open_memmap(f, mode='r') if mem_map else np.load(f)
            f = format_path(filename, width, i, block_diagonal, DecompositionFileType.data_npy)
            if os.path.exists(f):
                data = np.lib.format.open_memmap(f, mode='r') if mem_map else np.load(f)
            else:
                data = np.ones(indices.size, dtype=np.float32)
            if mem_map:
                B = (data, indices, indptr)
            else:
                B = sparse.csr_matrix((data, indices, indptr))
            if no_permutation:
                permutation = None
            else:
                f = format_path(filename, width, i, block_diagonal, DecompositionFileType.permutation_npy)
                permutation = np.load(f)
        except FileNotFoundError:
            # A part saved under a DIFFERENT width suffix (the reference
            # saves each part under its own arrow_width, graphio.py:176-183)
            # would be silently truncated here — warn instead of hiding it.
            import glob as _glob
            import warnings
            bd = "_bd" if block_diagonal else ""
            pattern = f"{filename}_B_*_{i}{bd}_indptr.npy"
            others = [p for p in _glob.glob(pattern)
                      if p != format_path(filename, width, i, block_diagonal,
                                          DecompositionFileType.indptr_npy)]
            if others:
                warnings.warn(
                    f"load_decomposition_new: stopping at part {i} for "
                    f"width={width}, but that part exists under a different "
                    f"width suffix: {sorted(others)} (per-part arrow widths "
                    f"— load with the matching width)")
            break
        decomposition.append((B, permutation))
        i += 1
    return decomposition


def load_decomposition(filename: str, width: Optional[int] = None,
                       block_diagonal: bool = True, no_permutation: bool = False):
    """Legacy .npz format (reference graphio.py:194-248): part i stored as
    {prefix}_B_{width}_{i}[_bd].npz + ..._permutation.npy."""
    decomposition = []
    i = 0
    while True:
        path = format_path(filename, width, i, block_diagonal, DecompositionFileType.npz)
        if not os.path.exists(path):
            break
        B = sparse.load_npz(path)
        if no_permutation:
            permutation = None
        else:
            p = format_path(filename, width, i, block_diagonal,
                            DecompositionFileType.permutation_npy)
            permutation = np.load(p)
        decomposition.append((B, permutation))
        i += 1
    return decomposition


def save_decomposition(decomposition, filename: str, width: int,
                       block_diagonal: bool = True) -> None:
    """Legacy .npz writer (reference graphio.py:103-117)."""
    for i, (B, permutation) in enumerate(decomposition):
        sparse.save_npz(format_path(filename, width, i, block_diagonal,
                                    DecompositionFileType.npz),
                        sparse.csr_matrix(B))
        np.save(format_path(filename, width, i, block_diagonal,
                            DecompositionFileType.permutation_npy),
                np.asarray(permutation))


def convert_decomposition(filename: str, width: Optional[int] = None,
                          block_diagonal: bool = True):
    """Convert a legacy .npz decomposition to the .npy CSR files
    (reference graphio.py:317-358)."""
    decomposition = load_decomposition(filename, width, block_diagonal,
                                       no_permutation=True)
    for i, (B, _) in enumerate(decomposition):
        B = sparse.csr_matrix(B)
        np.save(format_path(filename, width, i, block_diagonal,
                            DecompositionFileType.indptr_npy), B.indptr)
        np.save(format_path(filename, width, i, block_diagonal,
                            DecompositionFileType.indices_npy), B.indices)
        np.save(format_path(filename, width, i, block_diagonal,
                            DecompositionFileType.data_npy), B.data)
    return decomposition


def split_matrix_to_blocks(A: sparse.csr_matrix, block_size: int,
                           dtype=None, use_min_shape: bool = False
                           ) -> List[List[Union[sparse.csr_matrix, None]]]:
    """Split A into the arrow-pattern blocks.

    Semantics of reference graphio.py:361-406: for block-row i>0 only the
    blocks j in {0, i-1, i, i+1} are materialised (None elsewhere); block-row 0
    keeps every column block. A short last block-row is padded to block_size
    rows by edge-padding indptr AND declared square (block_size x block_size).
    Blocks are canonicalised (sum_duplicates + sort_indices).
    """
    rows, cols = A.shape
    dtype = dtype or A.dtype

    blocks_per_col = int(np.ceil(rows / block_size))
    blocks_per_row = int(np.ceil(cols / block_size))
    blocks: List[List[Union[sparse.csr_matrix, None]]] = \
        [[None] * blocks_per_row for _ in range(blocks_per_col)]
    for i in range(blocks_per_col):
        for j in range(blocks_per_row):
            if i > 0 and j not in (0, i - 1, i, i + 1):
                continue
            shape = (min(rows - i * block_size, block_size),
                     min(cols - j * block_size, block_size))
            sl = A[i * block_size:min(rows, (i + 1) * block_size),
                   j * block_size:min(cols, (j + 1) * block_size)]
            pad_width = block_size - shape[0]
            if use_min_shape or pad_width == 0:
                block = sparse.csr_matrix(sl, shape=shape, dtype=dtype)
            else:
                # Edge-pad indptr so the block has block_size rows
                # (reference graphio.py:394-399).
                indx_ptr = np.pad(sl.indptr, (0, pad_width), mode='edge')
                block = sparse.csr_matrix((sl.data, sl.indices, indx_ptr),
                                          shape=(block_size, block_size), dtype=dtype)
            block.sum_duplicates()
            block.sort_indices()
            blocks[i][j] = block
    return blocks

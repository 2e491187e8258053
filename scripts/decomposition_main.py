"""arrow_decompose CLI — flag-compatible with the reference
(scripts/decomposition_main.py:109-208), numpy/scipy-native (no igraph,
no julia). Reads `{dataset_dir}/{name}/{name}.mtx` (matrix market) or a
scipy `.npz`, decomposes, and writes the `.npy` CSR decomposition files
the engine (and the reference) consume."""
import argparse
import os
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from arrow_matrix_amd import graphio
from arrow_matrix_amd.decomposition import arrow_decomposition


def load_matrix(path: Path, fmt: str, directed: bool):
    from scipy import sparse
    if fmt == 'mtx':
        from scipy.io import mmread
        A = sparse.csr_matrix(mmread(str(path)))
    elif fmt == 'npz':
        A = sparse.load_npz(str(path)).tocsr()
    else:
        raise ValueError(f"unknown format {fmt} (mtx or npz; the reference's "
                         f"matlab path needs .mat files we do not ship)")
    if not directed:
        A = sparse.csr_matrix(A.maximum(A.T))
    A.data = np.ones_like(A.data, dtype=np.float32)  # adjacency semantics
    return A


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument('--width', type=int, default=5000000)
    parser.add_argument('--dataset_dir', type=str, default='.')
    parser.add_argument('--dataset_name', nargs='+', type=str, required=True)
    parser.add_argument('--format', type=str, default='mtx',
                        help="graph file format (mtx or npz)")
    parser.add_argument('--directed', type=bool, default=False)
    parser.add_argument('--levels', type=int, default=10,
                        help='max number of decomposition parts')
    parser.add_argument('--seed', type=int, default=0)
    args = parser.parse_args()

    rng = np.random.default_rng(args.seed)
    datasets_directory = Path(args.dataset_dir).expanduser()
    names = args.dataset_name if isinstance(args.dataset_name, list) else [args.dataset_name]

    for name in names:
        dataset_dir = datasets_directory / name
        f = dataset_dir / f"{name}.{args.format}"
        if not f.exists():
            raise ValueError(f"File {f} does not exist")
        print(f"Loading {name} ...", flush=True)
        A = load_matrix(f, args.format, args.directed)
        print(f"Decomposing {name} (n={A.shape[0]}, nnz={A.nnz}) "
              f"width={args.width} ...", flush=True)
        decomp = arrow_decomposition(A, arrow_width=args.width,
                                     max_number_of_levels=args.levels,
                                     block_diagonal=True, rng=rng)
        print(f"Successfully decomposed into {len(decomp)} matrices.")
        pairs = [(a.graph, a.permutation) for a in decomp]
        graphio.save_decomposition_new(pairs, str(dataset_dir / name),
                                       decomp[0].arrow_width)
        print(f"Saved under {dataset_dir / name}_B_{decomp[0].arrow_width}_*")


if __name__ == '__main__':
    main()

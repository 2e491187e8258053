"""spmm_15d CLI — flag-compatible with the reference
(scripts/spmm_15d_main.py:20-69), driving the 1.5D A-stationary baseline."""
import argparse
import math
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from arrow_matrix_amd.common import utils, wb_logging
from arrow_matrix_amd.comm import default_comm
from arrow_matrix_amd.spmm_15d import Spmm15D


def main():
    parser = argparse.ArgumentParser(description='SpMM 1.5D benchmark.')
    parser.add_argument('-d', '--dataset', nargs="?",
                        choices=['random', 'file'], default='random')
    parser.add_argument('-s', '--seed', type=int, nargs="?", default=42)
    parser.add_argument('-v', '--vertices', type=int, nargs="?", default=100000)
    parser.add_argument('-e', '--edges', type=int, nargs="?", default=1000000)
    parser.add_argument('-t', '--type', nargs="?",
                        choices=['float32', 'float64'], default='float32')
    parser.add_argument('-f', '--file', type=str, nargs="?", default=None,
                        help='scipy .npz file containing the sparse matrix')
    parser.add_argument('-c', '--columns', type=int, nargs="?", default=128)
    parser.add_argument('-r', '--replication', type=int, nargs="?", default=0,
                        help='replication factor c (0 = largest power of two '
                             'whose square is <= world size)')
    parser.add_argument('--validate', type=utils.str2bool, nargs="?", default=True)
    parser.add_argument('-i', '--device', type=str, default='gpu')
    parser.add_argument('-z', '--iterations', type=int, default=10)
    args = vars(parser.parse_args())

    if args['type'] != 'float32':
        raise NotImplementedError("fp32 only (DESIGN.md §next)")

    comm = default_comm()
    rng = np.random.default_rng(args['seed'])

    if args['replication'] == 0:
        n = math.floor(math.log2(max(comm.size, 1)) / 2)
        args['replication'] = 2 ** n
        utils.mpi_print(comm.rank, f"Using replication factor {args['replication']}")

    from scipy import sparse
    if args['dataset'] == 'file':
        A = sparse.load_npz(args['file']).tocsr().astype(np.float32)
    else:
        A = utils.generate_sparse_matrix(args['vertices'], args['vertices'],
                                         args['edges'], np.float32, rng)

    wb_logging.wandb_init(comm, args['file'], args['columns'],
                          args['iterations'], args['device'], "15D_v0.1_AMD", 0,
                          os.environ.get('WANDB_API_KEY'))

    eng = Spmm15D(comm, A, args['columns'], c=args['replication'],
                  device=args['device'])
    x0 = eng.x * eng.lNKb
    X_full = None
    for i in range(args['iterations']):
        X_full = 2 * rng.random((A.shape[1], args['columns']),
                                dtype=np.float32) - 1
        X_local = X_full[x0:min(A.shape[1], x0 + eng.lNKb)]
        wb_logging.set_iteration_data({"iteration": i})
        tic = time.perf_counter()
        Y = eng.spmm(X_local.copy())
        toc = time.perf_counter()
        wb_logging.log({"spmm_time": toc - tic})
        utils.mpi_print(comm.rank, f"Iteration {i} -- {toc - tic} s")

    if args['validate'] and comm.rank == 0 and comm.size == 1:
        ref = A @ X_full
        err = np.abs(Y.cpu().numpy() - ref[:Y.shape[0]]).max()
        print(f"VALIDATION max |err| = {err}")
        assert err < 1e-3 * max(1.0, abs(ref).max())

    wb_logging.finish()
    comm.barrier()


if __name__ == '__main__':
    main()

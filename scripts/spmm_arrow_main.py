"""spmm_arrow CLI — flag-compatible with the reference
(scripts/spmm_arrow_main.py:10-31; console script setup.py:20)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from arrow_matrix_amd import arrow_bench
from arrow_matrix_amd.comm import default_comm
from arrow_matrix_amd.common import utils


def main() -> None:
    # Flags and defaults are the reference CLI's (spmm_arrow_main.py:10-31);
    # help text reworded here.
    parser = argparse.ArgumentParser(description='Iterated arrow-SpMM benchmark')
    parser.add_argument('-f', '--path', type=str, default=None,
                        help='decomposed-graph file prefix; omit to run on '
                             'generated synthetic data')
    parser.add_argument('-w', '--width', type=int, default=0,
                        help='arrow width (block height) of the decomposition')
    parser.add_argument('-c', '--features', type=int, default=16,
                        help='feature columns k of the dense operand X')
    parser.add_argument('-b', '--blocked', type=utils.str2bool, nargs="?", default=True,
                        help='block-diagonal decomposition (False: banded '
                             'with +-1 off-diagonal blocks)')
    parser.add_argument('-i', '--device', type=str, default='gpu',
                        help="compute device: 'gpu' (HIP kernels) or 'cpu' "
                             "(scipy, the parity reference)")
    parser.add_argument('-z', '--iterations', type=int, default=1,
                        help='number of X <- A @ X iterations')
    parser.add_argument('-r', '--ranksperside', type=int, default=3,
                        help='synthetic data only: block-rows per matrix')
    parser.add_argument('-m', '--ba_neighbors', type=int, default=3,
                        help='synthetic data only: average neighbors per vertex')
    parser.add_argument('-s', '--slim', type=utils.str2bool, nargs="?", default=True,
                        help='slim layout (one rank per block-row); False '
                             'selects the banded ArrowMPI variant')
    parser.add_argument('-n', '--npy', type=utils.str2bool, nargs="?", default=True,
                        help='load the .npy (indptr/indices/data) format; '
                             'False loads the legacy .npz format')

    args = vars(parser.parse_args())
    comm = default_comm()
    utils.mpi_print(comm.rank, str(args))

    args['wandb_key'] = os.environ.get('WANDB_API_KEY')
    if args['wandb_key'] is None:
        utils.mpi_print(comm.rank,
                        "Set the WANDB_API_KEY environment variable to start "
                        "logging results to Weights & Biases.")

    arrow_bench.bench_spmm(args['path'],
                           args['width'],
                           args['features'],
                           args['iterations'],
                           args['blocked'],
                           args['device'],
                           args['ranksperside'],
                           args['ba_neighbors'],
                           args['wandb_key'],
                           slim=args['slim'],
                           npy_format=args['npy'])


if __name__ == '__main__':
    main()

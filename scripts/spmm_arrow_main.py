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
    parser = argparse.ArgumentParser(description='Benchmark the SpMM')
    parser.add_argument('-f', '--path', type=str, default=None,
                        help='The filename prefix of the decomposed graph. '
                             'If none, synthetic data is generated.')
    parser.add_argument('-w', '--width', type=int, default=0,
                        help='Width of the decomposition / Height of the blocks.')
    parser.add_argument('-c', '--features', type=int, default=16,
                        help='Width of the decomposition / Height of the blocks.')
    parser.add_argument('-b', '--blocked', type=utils.str2bool, nargs="?", default=True,
                        help='If true, the matrix has only one block diagonal,')
    parser.add_argument('-i', '--device', type=str, default='gpu',
                        help='Device to use for the MM. Either cpu or gpu.')
    parser.add_argument('-z', '--iterations', type=int, default=1,
                        help='Number of SpMM iteration to run.')
    parser.add_argument('-r', '--ranksperside', type=int, default=3,
                        help='Number of Ranks per Side (For synthetic data only)')
    parser.add_argument('-m', '--ba_neighbors', type=int, default=3,
                        help='Number of neighbors per bertex (For synthetic data only)')
    parser.add_argument('-s', '--slim', type=utils.str2bool, nargs="?", default=True,
                        help='If true, the decomposition onto ranks is "slim" '
                             'assigning one rank per row-block.')
    parser.add_argument('-n', '--npy', type=utils.str2bool, nargs="?", default=True,
                        help='If true, the decomposition is loaded from the '
                             'indices / indptr files.')

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

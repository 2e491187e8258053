"""spmm_petsc CLI — flag-compatible with the reference
(scripts/spmm_petsc_main.py)."""
import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from arrow_matrix_amd.common import utils
from arrow_matrix_amd.petsc_bench import benchmark_spmm


def main() -> None:
    parser = argparse.ArgumentParser(description='SpMM PETSc benchmark.')
    parser.add_argument('-s', '--seed', type=int, nargs="?", default=42)
    parser.add_argument('-t', '--type', nargs="?",
                        choices=['float32', 'float64'], default='float32')
    parser.add_argument('-f', '--file', type=str, nargs="?", default=None,
                        help='Matrix slice file of the form '
                             '{name}.part.{x}.slice.{y}.npz')
    parser.add_argument('-c', '--columns', type=int, nargs="?", default=32)
    parser.add_argument('-i', '--device', type=str, default='gpu')
    parser.add_argument('-z', '--iterations', type=int, default=3)
    parser.add_argument('--gpu-tiling', type=utils.str2bool, nargs="?", default=False)
    parser.add_argument('--dryrun', type=utils.str2bool, nargs="?", default=False)
    parser.add_argument('-m', '--memory', type=float, default=0.9)
    args = vars(parser.parse_args())

    args['wandb_key'] = os.environ.get('WANDB_API_KEY')
    rng = np.random.default_rng(args['seed'])
    dtype = np.float32 if args['type'] == 'float32' else np.float64
    benchmark_spmm(args['file'], args['columns'], args['iterations'],
                   args['device'], args['wandb_key'], dtype, rng,
                   gpu_tiling=args['gpu_tiling'], dryrun=args['dryrun'],
                   mem_fraction=args['memory'])


if __name__ == '__main__':
    main()

"""spmm_arrow CLI smoke (cpu device, synthetic data) — the reference's
bench_spmm flow end to end (arrow_bench.py:12-137)."""
import os
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_cli_synthetic_cpu():
    with tempfile.TemporaryDirectory() as td:
        r = subprocess.run(
            [sys.executable, os.path.join(REPO, 'scripts', 'spmm_arrow_main.py'),
             '--width', '20', '--features', '4', '--iterations', '2',
             '--device', 'cpu', '--ranksperside', '3', '--ba_neighbors', '4'],
            cwd=td, capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr + r.stdout
        assert 'Iteration 1' in r.stdout
        assert 'FAILED' not in r.stdout


def test_cli_slim_false_runs_arrow_mpi():
    """--slim False routes to the ArrowMPI engine (block-diagonal here)."""
    with tempfile.TemporaryDirectory() as td:
        r = subprocess.run(
            [sys.executable, os.path.join(REPO, 'scripts', 'spmm_arrow_main.py'),
             '--width', '12', '--features', '3', '--iterations', '1',
             '--device', 'cpu', '--slim', 'false', '--ranksperside', '3',
             '--ba_neighbors', '4'],
            cwd=td, capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr + r.stdout
        assert 'FAILED' not in r.stdout


def test_bench_spmm_importable_signature():
    from arrow_matrix_amd.arrow_bench import bench_spmm
    import inspect
    params = list(inspect.signature(bench_spmm).parameters)
    # the reference's signature (arrow_bench.py:12-23)
    assert params[:8] == ['path', 'width', 'n_features', 'iterations',
                          'blocked', 'device', 'p_per_side', 'ba_neighbors']


def test_arrow_decompose_cli_mtx():
    """arrow_decompose CLI on a small .mtx dataset; output loads back."""
    import numpy as np
    from scipy import sparse
    from scipy.io import mmwrite
    with tempfile.TemporaryDirectory() as td:
        ddir = os.path.join(td, 'toy')
        os.makedirs(ddir)
        rng = np.random.RandomState(0)
        A = sparse.random(60, 60, density=0.1, random_state=rng, format='csr')
        A = sparse.csr_matrix(A.maximum(A.T))
        mmwrite(os.path.join(ddir, 'toy.mtx'), A)
        r = subprocess.run(
            [sys.executable, os.path.join(REPO, 'scripts', 'decomposition_main.py'),
             '--width', '12', '--dataset_dir', td, '--dataset_name', 'toy',
             '--format', 'mtx'],
            capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr + r.stdout
        from arrow_matrix_amd import graphio
        loaded = graphio.load_decomposition_new(os.path.join(ddir, 'toy'), 12)
        assert len(loaded) >= 1
        # files feed the engine
        from arrow_matrix_amd.arrow_dec import ArrowDecompositionMPI
        blocks, nb, tp, tn = ArrowDecompositionMPI.load_decomposition_new(
            None, os.path.join(ddir, 'toy'), 12)
        assert int(nb[0]) >= 1

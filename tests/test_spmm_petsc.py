"""PETSc-style 1D SpMM engine (SpmmPETSc over MatrixSlice): parity with
scipy A_i @ X (the reference's golden, test_spmmPETSc.py:44-121)."""
import os
import socket

import numpy as np
import pytest
import torch.multiprocessing as mp
from scipy import sparse

from arrow_matrix_amd.matrix_slice import MatrixSlice
from arrow_matrix_amd.spmm_petsc import SpmmPETSc


def test_single_rank_identity():
    A = sparse.csr_matrix(sparse.eye(16, dtype=np.float32))
    ms = MatrixSlice.initialize(None, A)
    eng = SpmmPETSc(None, ms, device='cpu')
    rng = np.random.default_rng(0)
    X = rng.random((16, 4), dtype=np.float32)
    Y = eng.spmm(X)
    np.testing.assert_allclose(Y.numpy(), X, rtol=1e-6)


def test_single_rank_random_and_accumulate():
    rng = np.random.RandomState(1)
    A = sparse.csr_matrix(sparse.random(30, 30, density=0.2, random_state=rng,
                                        format='csr'), dtype=np.float32)
    ms = MatrixSlice.initialize(None, A)
    eng = SpmmPETSc(None, ms, device='cpu')
    X = rng.rand(30, 5).astype(np.float32)
    Y = eng.spmm(X)
    np.testing.assert_allclose(Y.numpy(), A @ X, rtol=1e-5, atol=1e-6)
    # accumulate semantics: Y passed in is incremented (spmm_petsc.py:196)
    Y2 = eng.spmm(X, Y)
    np.testing.assert_allclose(Y2.numpy(), 2 * (A @ X), rtol=1e-5, atol=1e-6)


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, port, q):
    import torch.distributed as dist
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    dist.init_process_group('gloo', rank=rank, world_size=2)
    try:
        from arrow_matrix_amd.comm import TorchDistComm
        comm = TorchDistComm()
        for sizes, density, seed in [((7, 5), 0.3, 0), ((4, 8), 0.0, 1)]:
            n = sum(sizes)
            rng = np.random.RandomState(seed)
            A_full = sparse.csr_matrix(
                sparse.random(n, n, density=density, random_state=rng,
                              format='csr'), dtype=np.float32)
            r0 = sum(sizes[:rank])
            A_i = sparse.csr_matrix(A_full[r0:r0 + sizes[rank]])
            ms = MatrixSlice.initialize(comm, A_i)
            eng = SpmmPETSc(comm, ms, device='cpu')
            x_full = rng.rand(n, 3).astype(np.float32)
            Y = eng.spmm(x_full[ms.start_col:ms.end_col])
            np.testing.assert_allclose(Y.numpy(), A_i @ x_full,
                                       rtol=1e-5, atol=1e-6)
        if rank == 0:
            q.put("ok")
    except Exception:
        import traceback
        if rank == 0:
            q.put("FAIL: " + traceback.format_exc())
        raise
    finally:
        dist.destroy_process_group()


def test_petsc_gloo_world2():
    port = _free_port()
    ctx = mp.get_context('spawn')
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"worker failed (exitcode {p.exitcode})"
    assert q.get(timeout=10) == "ok"


def test_benchmark_spmm_synthetic_cpu():
    from arrow_matrix_amd.petsc_bench import benchmark_spmm
    Y = benchmark_spmm(None, 4, 2, 'cpu', rng=np.random.default_rng(3))
    assert Y is not None and np.isfinite(Y.numpy()).all()


@pytest.mark.gpu
def test_petsc_gpu():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    rng = np.random.RandomState(5)
    A = sparse.csr_matrix(sparse.random(500, 500, density=0.05,
                                        random_state=rng, format='csr'),
                          dtype=np.float32)
    ms = MatrixSlice.initialize(None, A)
    eng = SpmmPETSc(None, ms, device='gpu')
    X = rng.rand(500, 32).astype(np.float32)
    Y = eng.spmm(X)
    np.testing.assert_allclose(Y.cpu().numpy(), A @ X, rtol=1e-4, atol=1e-4)


def test_benchmark_spmm_slice_files():
    """Slice-file loading with the reference's naming
    ({name}.part.{x}.slice.{y}.npz, spmm_petsc.py:82-102)."""
    import tempfile
    from arrow_matrix_amd.petsc_bench import benchmark_spmm, load_matrix_slice
    rng = np.random.RandomState(8)
    A = sparse.csr_matrix(sparse.random(12, 12, density=0.3, random_state=rng,
                                        format='csr'), dtype=np.float32)
    with tempfile.TemporaryDirectory() as td:
        f = os.path.join(td, 'g.part.1.slice.0.npz')
        sparse.save_npz(f, A)
        B = load_matrix_slice(os.path.join(td, 'g.part.1.slice.7.npz'), 0)
        assert (B != A).nnz == 0
        Y = benchmark_spmm(f, 3, 1, 'cpu', rng=np.random.default_rng(1))
        assert Y is not None and np.isfinite(Y.numpy()).all()

"""1.5D A-stationary baseline: parity with scipy A @ X
(comparison algorithm)."""
import os
import socket

import numpy as np
import pytest
import torch.multiprocessing as mp
from scipy import sparse

from arrow_matrix_amd.spmm_15d import Spmm15D


def _rand(n, m, density, seed):
    rs = np.random.RandomState(seed)
    return sparse.csr_matrix(sparse.random(n, m, density=density,
                                           random_state=rs, format='csr'),
                             dtype=np.float32)


def test_single_process():
    A = _rand(24, 24, 0.2, 0)
    eng = Spmm15D(None, A, X_cols=5, c=1, device='cpu')
    rng = np.random.default_rng(0)
    X = rng.random((24, 5), dtype=np.float32)
    Y = eng.spmm(X)
    np.testing.assert_allclose(Y.numpy(), A @ X, rtol=1e-5, atol=1e-6)


def test_bad_replication_factor():
    A = _rand(8, 8, 0.5, 1)
    with pytest.raises(ValueError):
        Spmm15D(None, A, X_cols=2, c=3, device='cpu')


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, port, world, c, q):
    import torch.distributed as dist
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    dist.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from arrow_matrix_amd.comm import TorchDistComm
        comm = TorchDistComm()
        n, k = (31 if c == 1 else 32), 4
        A = _rand(n, n, 0.25, 7)
        eng = Spmm15D(comm, A, X_cols=k, c=c, device='cpu')
        rng = np.random.default_rng(9)
        X_full = rng.random((n, k), dtype=np.float32)
        # my replicated X panel: bcast-rank x owns rows [x*lNKb, (x+1)*lNKb)
        x0 = min(n, eng.x * eng.lNKb)
        X_local = X_full[x0:min(n, x0 + eng.lNKb)]
        Y = eng.spmm(X_local.copy())
        ref = (A @ X_full)[eng.my_rows[0]:eng.my_rows[1]]
        np.testing.assert_allclose(Y.numpy(), ref, rtol=1e-4, atol=1e-5)
        if rank == 0:
            q.put("ok")
    except Exception:
        import traceback
        if rank == 0:
            q.put("FAIL: " + traceback.format_exc())
        raise
    finally:
        dist.destroy_process_group()


def test_single_process_ragged():
    # n not divisible by the grid: panel/sub-block clipping paths
    A = _rand(25, 25, 0.3, 5)
    eng = Spmm15D(None, A, X_cols=3, c=1, device='cpu')
    rng = np.random.default_rng(1)
    X = rng.random((25, 3), dtype=np.float32)
    Y = eng.spmm(X)
    np.testing.assert_allclose(Y.numpy(), A @ X, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("world,c", [(2, 1), (4, 2)])
def test_15d_gloo(world, c):
    port = _free_port()
    ctx = mp.get_context('spawn')
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, port, world, c, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker failed (exitcode {p.exitcode})"
    assert q.get(timeout=10) == "ok"

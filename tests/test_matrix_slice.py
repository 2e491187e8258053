"""MatrixSlice (kept PETSc-path loader API): table semantics on one rank and
across 2 gloo ranks, mirroring the reference's own cases
(test_spmmPETSc.py: identity KAT :95-121, unequal slices / density 0 :44-92)."""
import os
import socket

import numpy as np
import pytest
import torch.multiprocessing as mp
from scipy import sparse

from arrow_matrix_amd.matrix_slice import MatrixSlice


def test_single_rank_identity():
    A = sparse.eye(8, format='csr', dtype=np.float32)
    ms = MatrixSlice.initialize(None, sparse.csr_matrix(A))
    assert ms.start_col == 0 and ms.end_col == 8
    assert ms.x_index_in.size == 0
    assert ms.A_i_nonlocal.shape == (8, 0)
    np.testing.assert_array_equal(ms.all_n_i, [8])


def test_single_rank_random():
    rng = np.random.RandomState(0)
    A = sparse.csr_matrix(sparse.random(10, 10, density=0.4, random_state=rng,
                                        format='csr'), dtype=np.float32)
    ms = MatrixSlice.initialize(None, A)
    assert ms.x_index_in.size == 0  # every column is local
    x = rng.rand(10, 3).astype(np.float32)
    y = ms.A_i_local @ x
    np.testing.assert_allclose(y, A @ x, rtol=1e-6)


def test_non_square_raises():
    A = sparse.csr_matrix(np.ones((4, 6), dtype=np.float32))
    with pytest.raises(ValueError):
        MatrixSlice.initialize(None, A)


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
        # unequal slices: 6 + 4 rows; plus a density-0 slice case
        for sizes, density, seed in [((6, 4), 0.4, 0), ((3, 7), 0.0, 1),
                                     ((5, 5), 0.15, 2)]:
            n = sum(sizes)
            rng = np.random.RandomState(seed)
            A_full = sparse.csr_matrix(
                sparse.random(n, n, density=density, random_state=rng,
                              format='csr'), dtype=np.float32)
            r0 = sum(sizes[:rank])
            A_i = sparse.csr_matrix(A_full[r0:r0 + sizes[rank]])
            ms = MatrixSlice.initialize(comm, A_i)
            np.testing.assert_array_equal(ms.all_n_i, sizes)
            # nonlocal columns are exactly A_i's out-of-slice columns
            mask = np.zeros(n, bool)
            mask[A_i.nonzero()[1]] = True
            mask[ms.start_col:ms.end_col] = False
            np.testing.assert_array_equal(ms.x_index_in, np.flatnonzero(mask))
            # SpMM through the tables: exchange x rows, then
            # y = A_local @ x_local + A_nonlocal @ x_in == (A_i @ x_full)
            x_full = rng.rand(n, 2).astype(np.float32)
            x_local = x_full[ms.start_col:ms.end_col]
            import torch
            send = torch.from_numpy(
                np.ascontiguousarray(x_local[ms.x_index_out_localized]))
            recv = comm.alltoallv(send, [int(c) for c in ms.send_count],
                                  [int(c) for c in ms.recv_count]).numpy()
            y = ms.A_i_local @ x_local + ms.A_i_nonlocal @ recv
            np.testing.assert_allclose(y, (A_i @ x_full), rtol=1e-5, atol=1e-6)
        if rank == 0:
            q.put("ok")
    except Exception:
        import traceback
        if rank == 0:
            q.put("FAIL: " + traceback.format_exc())
        raise
    finally:
        dist.destroy_process_group()


def test_matrix_slice_gloo_world2():
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
    res = q.get(timeout=10)
    assert res == "ok", res


def test_single_rank_vs_reference_golden():
    """Field-by-field pin against the REFERENCE MatrixSlice run at P=1
    (fixtures from tests/golden/gen_golden.py)."""
    golden = np.load(os.path.join(os.path.dirname(__file__), 'golden',
                                  'reference_katsets.npz'))
    n = int(golden['ms_n_cases'][0])
    assert n >= 4
    for ci in range(n):
        A = sparse.csr_matrix(golden[f'ms_{ci}_dense'].astype(np.float32))
        ms = MatrixSlice.initialize(None, A.copy())
        np.testing.assert_array_equal(ms.A_i_local.toarray(),
                                      golden[f'ms_{ci}_local'])
        for f in ('x_index_in', 'rank_in', 'x_index_out', 'rank_out',
                  'send_count', 'recv_count'):
            np.testing.assert_array_equal(np.asarray(getattr(ms, f)),
                                          golden[f'ms_{ci}_{f}'])
        np.testing.assert_array_equal([ms.start_col, ms.end_col],
                                      golden[f'ms_{ci}_bounds'])

"""Golden-vector generator — RUNS ONLY IN THE BUILD CONTAINER.

Imports the reference implementation (/root/reference, PUBLIC UNTRUSTED
CONTENT — we execute only its pure table/loader functions on our own inputs)
with mpi4py/igraph/tqdm stubbed, calls the pure functions of the hot path,
and commits their outputs as .npz fixtures so parity tests keep running on
machines where /root/reference does not exist (the GPU box).

Pinned functions:
  - ArrowDecompositionMPI._all_to_all_tables / _aggregation_permutation
    (arrow_dec_mpi.py:325-384)
  - ArrowDecompositionMPI.number_of_blocks (arrow_dec_mpi.py:612-627)
  - graphio.split_matrix_to_blocks (graphio.py:361-406)
  - ArrowDecompositionMPI.load_decomposition_new rank-0 path incl. the
    permutation pad/compose/sentinel rules and the block-scatter send
    protocol (arrow_dec_mpi.py:629-930), captured through a recording
    FakeComm.

Usage:  python tests/golden/gen_golden.py
"""
import os
import sys
import types

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.abspath(os.path.join(HERE, '..', '..'))
REFERENCE = os.environ.get('ARROW_REFERENCE_PATH', '/root/reference')


def _install_stubs():
    mpi4py = types.ModuleType('mpi4py')
    MPI = types.ModuleType('mpi4py.MPI')

    class _Dummy:  # placeholder for type annotations / attribute access
        pass

    MPI.Request = _Dummy
    MPI.Comm = _Dummy
    MPI.Group = _Dummy
    MPI.FLOAT = object()
    MPI.INT64_T = object()
    MPI.SUM = object()
    MPI.LOR = object()
    MPI.COMM_WORLD = None
    mpi4py.MPI = MPI
    sys.modules.setdefault('mpi4py', mpi4py)
    sys.modules.setdefault('mpi4py.MPI', MPI)
    for name in ('igraph', 'tqdm', 'wandb', 'cupy'):
        mod = types.ModuleType(name)
        if name == 'igraph':
            mod.Graph = _Dummy
        if name == 'tqdm':
            mod.tqdm = lambda x, *a, **kw: x
        sys.modules.setdefault(name, mod)


class FakeComm:
    """Records rank-0's Send traffic of the reference block scatter."""

    def __init__(self, size):
        self._size = size
        self.sends = []  # list of (dest, tag, array copy)

    def Get_rank(self):
        return 0

    def Get_size(self):
        return self._size

    rank = 0

    def Bcast(self, buf, root=0):
        pass  # rank 0 is the broadcaster; in-place no-op

    def Send(self, buf, dest, tag=0):
        self.sends.append((int(dest), int(tag), np.array(buf, copy=True)))


def main():
    _install_stubs()
    sys.path.insert(0, REFERENCE)
    sys.path.insert(0, REPO)
    from arrow import arrow_dec_mpi  # noqa: reference import (stubbed deps)
    from arrow.common import graphio as ref_graphio
    ADM = arrow_dec_mpi.ArrowDecompositionMPI

    from arrow_matrix_amd import graphio as my_graphio
    from arrow_matrix_amd import synth

    rng = np.random.default_rng(7)
    out = {}

    # ---- 1. all_to_all_tables KATs ----------------------------------------
    cases = []
    # reference's own unit-test shapes (test_arrowmpi.py:24-94)
    rpr, cols, ranks, prev_ranks = 4, 6, 2, 6
    perm = np.asarray(list(reversed(range(ranks * rpr))))
    for i in range(ranks):
        cases.append((perm[i * rpr:(i + 1) * rpr], rpr, cols, prev_ranks + ranks, prev_ranks))
        cases.append((perm[i * rpr:(i + 1) * rpr], rpr, cols, prev_ranks + ranks, 0))
    # random permutations with sentinel overflow values
    for t in range(8):
        rpr = int(rng.integers(3, 40))
        total = int(rng.integers(2, 9))
        off = int(rng.integers(0, total))
        vals = rng.integers(0, rpr * (total + 2), size=rpr)  # some out of range
        if t % 2 == 0:
            vals[rng.integers(0, rpr, size=max(1, rpr // 4))] = 2 * rpr * total  # sentinel
        cases.append((vals.astype(np.int64), rpr, int(rng.integers(1, 20)), total, off))
    for ci, (p, r, c, t, o) in enumerate(cases):
        counts, displs, sp, ap = ADM._all_to_all_tables(np.asarray(p), r, c, t, o)
        out[f'a2a_{ci}_in_perm'] = np.asarray(p)
        out[f'a2a_{ci}_meta'] = np.asarray([r, c, t, o])
        out[f'a2a_{ci}_counts'] = np.asarray(counts)
        out[f'a2a_{ci}_displs'] = np.asarray(displs)
        out[f'a2a_{ci}_send_perm'] = np.asarray(sp)
        out[f'a2a_{ci}_agg_perm'] = np.asarray(ap)
    out['a2a_n_cases'] = np.asarray([len(cases)])

    # ---- 2. number_of_blocks KATs -----------------------------------------
    from scipy import sparse
    nb_cases = []
    for t in range(6):
        n = int(rng.integers(10, 200))
        w = int(rng.integers(2, 30))
        nnz_rows = int(rng.integers(1, n + 1))
        m = sparse.random(n, n, density=0.05, format='csr', random_state=np.random.RandomState(t))
        m = sparse.csr_matrix(m)
        # zero out trailing rows
        dense = m.toarray()
        dense[nnz_rows:, :] = 0
        if dense[:nnz_rows].sum() == 0:
            dense[nnz_rows - 1, 0] = 1.0
        m = sparse.csr_matrix(dense)
        nb_cases.append((m, w))
    for ci, (m, w) in enumerate(nb_cases):
        out[f'nb_{ci}_nnz_per_row'] = np.asarray(m.getnnz(1))
        out[f'nb_{ci}_width'] = np.asarray([w])
        out[f'nb_{ci}_result'] = np.asarray([ADM.number_of_blocks(m, w)])
    out['nb_n_cases'] = np.asarray([len(nb_cases)])

    # ---- 3. split_matrix_to_blocks KATs -----------------------------------
    sp_cases = []
    for t, (n, bs) in enumerate([(12, 4), (13, 4), (30, 7), (8, 8), (9, 2)]):
        m = sparse.csr_matrix(sparse.random(n, n, density=0.3,
                                            random_state=np.random.RandomState(10 + t),
                                            format='csr'), dtype=np.float32)
        sp_cases.append((m, bs))
    for ci, (m, bs) in enumerate(sp_cases):
        blocks = ref_graphio.split_matrix_to_blocks(m, bs)
        out[f'split_{ci}_dense'] = m.toarray()
        out[f'split_{ci}_bs'] = np.asarray([bs])
        out[f'split_{ci}_grid'] = np.asarray([len(blocks), len(blocks[0])])
        for i in range(len(blocks)):
            for j in range(len(blocks[0])):
                if blocks[i][j] is not None:
                    out[f'split_{ci}_blk_{i}_{j}'] = blocks[i][j].toarray()
                    out[f'split_{ci}_blk_{i}_{j}_shape'] = np.asarray(blocks[i][j].shape)
    out['split_n_cases'] = np.asarray([len(sp_cases)])

    # ---- 4. loader distribution capture -----------------------------------
    # Synthetic decompositions written in the on-disk format by OUR graphio,
    # loaded+scattered by the REFERENCE loader with a recording FakeComm.
    import tempfile
    loader_cases = [
        dict(width=6, n_blocks=[3], seed=11, one_based=False, banded=False),
        dict(width=5, n_blocks=[4, 2], seed=12, one_based=False, banded=False),
        dict(width=4, n_blocks=[4, 3, 2], seed=13, one_based=False, banded=False),
        dict(width=6, n_blocks=[2, 2], seed=14, one_based=True, banded=False),
        dict(width=5, n_blocks=[4], seed=15, one_based=False, banded=True),
        dict(width=4, n_blocks=[3, 3], seed=16, one_based=False, banded=True),
    ]
    for ci, cfg in enumerate(loader_cases):
        banded = cfg.get('banded', False)
        decomp = synth.synth_arrow_decomposition(cfg['width'], cfg['n_blocks'],
                                                 avg_deg=4, seed=cfg['seed'],
                                                 block_diagonal=not banded)
        if cfg['one_based']:
            decomp = [(B, p + 1) for B, p in decomp]
        with tempfile.TemporaryDirectory() as td:
            prefix = os.path.join(td, 'g')
            my_graphio.save_decomposition_new(decomp, prefix, cfg['width'],
                                              block_diagonal=not banded)
            total = (sum(cfg['n_blocks']) if not banded
                     else sum(2 * b - 1 for b in cfg['n_blocks']))
            comm = FakeComm(total)
            blocks0, n_blocks, to_prev0, to_next0 = ADM.load_decomposition_new(
                comm, prefix, cfg['width'], is_block_diagonal=not banded,
                slim=not banded, use_npy=True, use_mmap=False)
        out[f'load_{ci}_n_blocks'] = np.asarray(n_blocks)
        out[f'load_{ci}_meta'] = np.asarray([cfg['width'], len(cfg['n_blocks']),
                                             int(cfg['one_based']), cfg['seed'],
                                             int(banded)])
        if to_next0 is not None:
            out[f'load_{ci}_rank0_to_next'] = to_next0
        # rank-0's own blocks (row 0 of part 0: A_00 only in slim scatter)
        for bi in range(len(blocks0)):
            for bj in range(len(blocks0[0])):
                if blocks0[bi][bj] is not None:
                    out[f'load_{ci}_rank0_blk_{bi}_{bj}'] = blocks0[bi][bj].toarray()
        # captured send stream: (dest, tag) -> payload sequence
        for si, (dest, tag, buf) in enumerate(comm.sends):
            out[f'load_{ci}_send_{si}_hdr'] = np.asarray([dest, tag])
            out[f'load_{ci}_send_{si}_buf'] = buf
        out[f'load_{ci}_n_sends'] = np.asarray([len(comm.sends)])
    out['load_n_cases'] = np.asarray([len(loader_cases)])

    # ---- 5. MatrixSlice single-rank golden --------------------------------
    # (matrix_slice.py:106-154; P=1 makes the MPI exchanges identities we
    # can fake, pinning the table construction + localisation fields)
    class FakeComm1:
        def Get_rank(self):
            return 0

        def Get_size(self):
            return 1

        def allgather(self, x):
            return [x]

        def alltoall(self, xs):
            return list(xs)

        def Alltoall(self, send, recv):
            np.copyto(recv, np.asarray(send))

        def Alltoallv(self, send_spec, recv_spec):
            recv_spec[0][:send_spec[0].size] = send_spec[0]

        def Barrier(self):
            pass

    from arrow import matrix_slice as ref_ms
    ms_cases = []
    for t in range(4):
        nn = int(rng.integers(6, 40))
        dens = float(rng.random() * 0.4)
        m = sparse.csr_matrix(sparse.random(nn, nn, density=dens,
                                            random_state=np.random.RandomState(50 + t),
                                            format='csr'), dtype=np.float32)
        ms_cases.append(m)
    for ci, m in enumerate(ms_cases):
        ms = ref_ms.MatrixSlice.initialize(FakeComm1(), m.copy())
        out[f'ms_{ci}_dense'] = m.toarray()
        out[f'ms_{ci}_local'] = ms.A_i_local.toarray()
        out[f'ms_{ci}_x_index_in'] = np.asarray(ms.x_index_in)
        out[f'ms_{ci}_rank_in'] = np.asarray(ms.rank_in)
        out[f'ms_{ci}_x_index_out'] = np.asarray(ms.x_index_out)
        out[f'ms_{ci}_rank_out'] = np.asarray(ms.rank_out)
        out[f'ms_{ci}_send_count'] = np.asarray(ms.send_count)
        out[f'ms_{ci}_recv_count'] = np.asarray(ms.recv_count)
        out[f'ms_{ci}_bounds'] = np.asarray([ms.start_col, ms.end_col])
    out['ms_n_cases'] = np.asarray([len(ms_cases)])

    path = os.environ.get('ARROW_GOLDEN_OUT',
                          os.path.join(HERE, 'reference_katsets.npz'))
    np.savez_compressed(path, **out)
    print(f'wrote {path} ({os.path.getsize(path)/1024:.1f} KiB, {len(out)} arrays)')


if __name__ == '__main__':
    main()

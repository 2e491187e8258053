"""MatrixSlice — the kept 1D row-slice loader API of the PETSc-style path.

Re-implements the reference's `arrow/matrix_slice.py` (constructor fields
:11-80, `initialize` :106-154, receive tables :184-227, send tables
:229-273) without mpi4py: the collectives go through comm.py (gloo on CPU,
RCCL on GPU). The table construction is vectorised (searchsorted instead of
the rank scan loop) but produces identical arrays — same sort orders, same
tie-breaking (sorted by (rank, index), matrix_slice.py:266-267).

The PETSc-style SpMM itself (spmm_petsc.py) is a "next" row
(SURVEY.md §8f-2); this class is its loader/comm-table surface, kept
importable and working on CPU as north_star requires.
"""
from typing import Optional

import numpy as np

from .comm import Comm


class MatrixSlice:

    def __init__(self, A_i_local, A_i_nonlocal, x_index_in: np.ndarray,
                 rank_in: np.ndarray, x_index_out: np.ndarray,
                 rank_out: np.ndarray, all_n_i: np.ndarray, start_col: int,
                 end_col: int, send_count: np.ndarray, recv_count: np.ndarray):
        # invariants of the reference constructor (matrix_slice.py:36-49)
        assert A_i_local.shape[0] == A_i_nonlocal.shape[0]
        assert A_i_local.shape[0] == A_i_local.shape[1]
        assert x_index_in.shape[0] == rank_in.shape[0]
        assert x_index_out.shape[0] == rank_out.shape[0]
        assert x_index_in.shape[0] == A_i_nonlocal.shape[1]
        assert MatrixSlice._is_sorted(x_index_in)
        assert MatrixSlice._is_sorted(rank_out)
        assert MatrixSlice._is_sorted(rank_in)

        self.A_i_local = A_i_local
        self.A_i_nonlocal = A_i_nonlocal
        self.x_index_in = x_index_in
        self.rank_in = rank_in
        self.x_index_out = x_index_out
        self.rank_out = rank_out
        self.all_n_i = all_n_i
        self.start_col = start_col
        self.end_col = end_col

        # local indices into the local X_i (matrix_slice.py:63-72)
        self.x_index_out_localized = x_index_out - start_col
        start_col_all_ranks = np.insert(np.cumsum(all_n_i), 0, 0)
        self.x_index_in_localized = x_index_in - start_col_all_ranks[rank_in]

        self.send_count = send_count
        self.recv_count = recv_count
        self.send_sdispl = np.insert(np.cumsum(send_count), 0, 0)
        self.recv_sdispl = np.insert(np.cumsum(recv_count), 0, 0)

    @staticmethod
    def _is_sorted(a):
        a = np.asarray(a)
        return a.size < 2 or bool(np.all(a[:-1] <= a[1:]))

    @staticmethod
    def get_local_matrix_dimensions(comm: Comm, A_i) -> np.ndarray:
        """All-gather each rank's row count (matrix_slice.py:82-92)."""
        return np.asarray(comm.allgather_int(A_i.shape[0]))

    @staticmethod
    def identify_local_slice(rank: int, all_n_i) -> tuple:
        start_col = int(np.sum(all_n_i[:rank]))
        end_col = int(np.sum(all_n_i[:rank + 1]))
        return start_col, end_col

    @classmethod
    def initialize(cls, comm: Optional[Comm], A_i) -> 'MatrixSlice':
        """Build the communication tables (matrix_slice.py:106-154)."""
        comm = comm if comm is not None else Comm()
        rank = comm.rank

        all_n_i = cls.get_local_matrix_dimensions(comm, A_i)
        total_rows = int(np.sum(all_n_i))
        if total_rows != A_i.shape[1]:
            raise ValueError(
                f"Matrix not square: Rank {rank} has {A_i.shape[1]} columns, "
                f"but the total number of rows is {total_rows}")

        start_col, end_col = cls.identify_local_slice(rank, all_n_i)

        A_i_local = A_i[:, start_col:end_col]
        A_i_local.sort_indices()
        A_i_local.sum_duplicates()
        A_i_local.eliminate_zeros()

        non_local_columns, x_index_in, rank_in = cls.construct_receive_tables(
            A_i, start_col, end_col, all_n_i)
        recv_counts = np.bincount(rank_in, minlength=comm.size)
        x_index_out, rank_out, send_counts = cls.construct_send_tables(
            comm, rank_in, x_index_in, recv_counts)
        assert cls.check_comm_tables(comm, x_index_in, rank_in, x_index_out, rank_out)

        A_i_nonlocal = A_i[:, non_local_columns]
        assert A_i_nonlocal.shape[1] == len(x_index_in)
        comm.barrier()
        return cls(A_i_local, A_i_nonlocal, x_index_in, rank_in, x_index_out,
                   rank_out, all_n_i, start_col, end_col, send_counts, recv_counts)

    @staticmethod
    def construct_receive_tables(A_i, start_col: int, end_col: int,
                                 all_n_i: np.ndarray):
        """Non-local column table (matrix_slice.py:184-227), vectorised:
        rank of column c = first r with c < cumsum(all_n_i)[r]."""
        first_slice = A_i[:, :start_col].nonzero()[1]
        second_slice = A_i[:, end_col:].nonzero()[1] + end_col
        nonlocal_cols = np.unique(np.concatenate((first_slice, second_slice)))
        cumulative = np.cumsum(all_n_i)
        rank_in = np.searchsorted(cumulative, nonlocal_cols, side='right').astype(np.int64)
        x_index_in = nonlocal_cols.astype(np.int64)
        return nonlocal_cols, x_index_in, rank_in

    @staticmethod
    def construct_send_tables(comm: Comm, rank_in: np.ndarray,
                              x_index_in: np.ndarray, recv_counts: np.ndarray):
        """Exchange the receive wishes (matrix_slice.py:234-273): alltoall of
        counts, then alltoallv of the requested indices; output sorted by
        (rank, index)."""
        import torch
        send_counts = np.asarray(comm.alltoall_ints([int(c) for c in recv_counts]),
                                 dtype=np.int64)
        send_buffer = comm.alltoallv(
            torch.from_numpy(x_index_in.astype(np.int64)).view(-1, 1),
            [int(c) for c in recv_counts], [int(c) for c in send_counts]
        ).view(-1).numpy()
        rank_out = np.repeat(np.arange(comm.size, dtype=np.int64), send_counts)
        # sort by (rank, index) (matrix_slice.py:267); each rank's segment is
        # already grouped — sort indices within each segment
        x_index_out = send_buffer.copy()
        order = np.lexsort((x_index_out, rank_out))
        return x_index_out[order], rank_out[order], send_counts

    @staticmethod
    def check_comm_tables(comm: Comm, x_index_in, rank_in, x_index_out, rank_out) -> bool:
        """Count symmetry check (matrix_slice.py:157-182)."""
        send_counts = np.bincount(rank_out, minlength=comm.size)
        recv_counts = np.bincount(rank_in, minlength=comm.size)
        recv_counts_c = comm.alltoall_ints([int(c) for c in send_counts])
        send_counts_c = comm.alltoall_ints([int(c) for c in recv_counts])
        ok = (list(send_counts) == list(send_counts_c)
              and list(recv_counts) == list(recv_counts_c))
        assert ok
        return ok

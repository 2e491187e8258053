"""A-stationary 1.5D SpMM baseline (comparison point).

Re-implements the semantics of the reference's
`arrow/baseline/spmm_15d.py`: a P/c × c process grid; rank (x, y) holds the
(row-panel x, column-panel y) block of A, split into `rounds = P/c²` column
sub-blocks; per round the owning rank broadcasts its X panel within the
column subcommunicator and every rank accumulates Y += A_r @ X_panel
(spmm_15d_cpu, :312-370); finally Y is allreduced over the replication
subcommunicator.

Deviations (same spirit as the rest of this repo):
  * every rank slices its own A panels locally (shared filesystem /
    caller-provided matrix) instead of the reference's root scatter
    (:84-119);
  * the multiplies run on the resident HIP kernel (device='gpu') or scipy
    (device='cpu'); collectives are RCCL/gloo via torch.distributed
    subgroups instead of MPI cartesian communicators.
"""
from typing import List, Optional

import numpy as np
import torch
from scipy import sparse

from .backends import make_backend
from .comm import Comm, TorchDistComm
from .common import wb_logging
import time

try:
    import torch.distributed as dist
except Exception:  # pragma: no cover
    dist = None


class Spmm15D:

    def __init__(self, comm: Optional[Comm], A: sparse.csr_matrix, X_cols: int,
                 c: int = 1, device: str = 'cpu'):
        """:param A: the FULL matrix (every rank slices its own panels).
        :param c: replication factor; world size must equal (P/c)·c with
        P/c divisible by c (reference :35-41)."""
        self.comm = comm if comm is not None else Comm()
        P = self.comm.size
        if P % c != 0:
            raise ValueError("The number of processes must be divisible by "
                             "the replication factor.")
        p_div_c = P // c
        if (p_div_c // c) * c != p_div_c:
            raise ValueError("The number of processes must be divisible by "
                             "the square of the replication factor.")
        self.c = c
        self.p_div_c = p_div_c
        self.rounds = p_div_c // c
        rank = self.comm.rank
        self.x, self.y = rank // c, rank % c  # row-major cartesian coords

        self.backend = make_backend(device)
        A = sparse.csr_matrix(A)
        NI, NK = A.shape
        self.NK = NK
        self.NJ = X_cols
        self.lNI = int(np.ceil(NI / p_div_c))
        self.lNKb = int(np.ceil(NK / p_div_c))
        # panel width is rounds*lNKb (reference :81 redefines lNK this way)
        # so sub-block boundaries line up with the bcast roots' X blocks on
        # ragged sizes
        lNK_panel = self.rounds * self.lNKb

        # my A panel, split into `rounds` column sub-blocks (:124-135)
        r0, r1 = self.x * self.lNI, min(NI, (self.x + 1) * self.lNI)
        c0, c1 = min(NK, self.y * lNK_panel), min(NK, (self.y + 1) * lNK_panel)
        panel = sparse.csr_matrix(A[r0:r1, c0:c1])
        self.my_rows = (r0, r1)
        self.A_blocks: List = []
        self.block_cols: List[int] = []
        for i in range(self.rounds):
            b0, b1 = i * self.lNKb, min(panel.shape[1], (i + 1) * self.lNKb)
            blk = sparse.csr_matrix(panel[:, b0:max(b1, b0)])
            self.A_blocks.append(self.backend.upload_block(blk))
            self.block_cols.append(blk.shape[1])

        # subcommunicators: bcast over the column (same y), reduce over the
        # replicas (same x) — reference cart_comm.Sub calls (:52-66)
        self.bcast_comm: Comm = Comm()
        self.reduce_comm: Comm = Comm()
        if P > 1:
            assert dist is not None and dist.is_initialized()
            for j in range(c):
                ranks = [xx * c + j for xx in range(p_div_c)]
                g = dist.new_group(ranks)
                if j == self.y:
                    self.bcast_comm = TorchDistComm(g)
            for xx in range(p_div_c):
                ranks = [xx * c + yy for yy in range(c)]
                g = dist.new_group(ranks)
                if xx == self.x:
                    self.reduce_comm = TorchDistComm(g)

    def x_block_rows(self) -> int:
        """Rows of the X block this rank broadcasts (block index = x
        coordinate): min(NK, (x+1)*lNKb) - x*lNKb, which differs from the
        rank's own first A sub-block width on ragged NK."""
        if not self.rounds:
            return 0
        lo = min(self.NK, self.x * self.lNKb)
        return min(self.NK, (self.x + 1) * self.lNKb) - lo

    def spmm(self, X_local: torch.Tensor) -> torch.Tensor:
        """One Y = A @ X round sweep (reference spmm_15d_cpu :312-370).
        X_local: this rank's X panel (lNKb rows, replicated across the
        reduce communicator). Returns this rank's Y row panel."""
        be = self.backend
        X_local = be.asarray(X_local)
        k = X_local.shape[1]
        Y = be.zeros((self.my_rows[1] - self.my_rows[0], k))

        bcast_t = kernel_t = 0.0
        for r in range(self.rounds):
            q = self.y * self.rounds + r  # owning bcast group-rank (:336)
            tic = time.perf_counter()
            if self.bcast_comm.rank == q:
                buf = X_local.contiguous()
            else:
                buf = be.zeros((self.block_cols[r], k))
            # torch.distributed broadcast takes the GLOBAL rank of the root;
            # group-index q in column-y's bcast group is global rank q*c + y
            self.bcast_comm.bcast_(buf, src=q * self.c + self.y)
            bcast_t += time.perf_counter() - tic

            tic = time.perf_counter()
            if self.block_cols[r] > 0:
                be.spmm_block(self.A_blocks[r], buf, Y, 1)
            kernel_t += time.perf_counter() - tic

        tic = time.perf_counter()
        self.reduce_comm.allreduce_sum_(Y)
        wb_logging.log({"spmm_bcast_time": bcast_t})
        wb_logging.log({"spmm_kernel_time": kernel_t})
        wb_logging.log({"spmm_reduce_time": time.perf_counter() - tic})
        return Y

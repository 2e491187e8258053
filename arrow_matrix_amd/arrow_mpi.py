"""ArrowMPI — the split/banded arrow variant (`--slim False`).

Re-implements the semantics of the reference's `arrow/arrow_mpi.py`: on top
of the slim dataflow, a banded (non-block-diagonal) decomposition carries
the off-diagonal blocks A_{r,r-1}/A_{r,r+1} whose X tiles come from the ±1
neighbours (halo Isend/Irecv, arrow_mpi.py:123-175; column-tile compute
arrow_mpi.py:177-219).

MI355X-first deviation: the reference splits each
matrix over 2b-1 ranks (b row ranks + b-1 column ranks,
arrow_mpi.py:338-344) because its CPU ranks are compute-bound; here the
same block-rows live on 1..8 GPUs like the slim engine (one GPU computes
both the row and column tile of its blocks), the halo exchange is grouped
p2p over xGMI, and the interior off-diagonals are folded into the fused
resident structure. Every rank is a column rank (holds feature tiles).
"""
from typing import Optional

from .arrow_slim import ArrowSlimMPI
from .comm import Comm


class ArrowMPI(ArrowSlimMPI):

    def __init__(self, comm: Optional[Comm] = None,
                 is_block_diagonal: bool = False,
                 tiles_per_side: Optional[int] = None, device: str = 'cpu'):
        super().__init__(comm, tiles_per_side=tiles_per_side, device=device,
                         banded=not is_block_diagonal)
        self.is_block_diagonal = is_block_diagonal

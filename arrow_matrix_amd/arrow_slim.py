"""Slim arrow SpMM engine — the per-matrix compute of the hot path.

Re-implements the semantics of the reference's `ArrowSlimMPI`
(arrow_slim_mpi.py): block-row i of an arrow matrix holds A_0i (first
block-row), A_ii (diagonal) and A_i0 (first block-column); per iteration

    C_0 = sum_i A_0i @ X_i            (reduced to the owner of block-row 0,
                                       arrow_slim_mpi.py:104-116)
    C_i = A_ii @ X_i + A_i0 @ X_0     (arrow_slim_mpi.py:121-144)

with X_0 broadcast (arrow_slim_mpi.py:269-273).

MI355X-first deviations from the reference:
  * A rank owns a CONTIGUOUS SPAN of block-rows, not exactly one: 288 GB of
    HBM3E per GPU holds many width-wide tiles, so 1..8 ranks cover any
    decomposition (the reference needs sum(n_blocks) MPI ranks). At one
    block per rank this reduces to the reference layout.
  * Partial sums of C_0 over the rank's own blocks are accumulated on-GPU
    first; ONE reduce follows (the reference reduces one partial per rank).
  * A is uploaded to HBM once at load time and stays resident (the reference
    re-uploads every call, arrow_slim_mpi.py:184-232 — noted TODO at
    arrow_mpi.py:314).
  * Collectives are RCCL over xGMI (comm.py) instead of mpi4py.
"""
from typing import List, Optional

import numpy as np
import torch

from .arrow_matrix import ArrowMatrix
from .backends import make_backend
from .comm import Comm
from .common import wb_logging
import time


class ArrowSlimMPI(ArrowMatrix):

    def __init__(self, comm: Optional[Comm] = None, tiles_per_side: Optional[int] = None,
                 device: str = 'cpu', banded: bool = False):
        """:param comm: communicator over this matrix's ranks (None = single
        process). :param tiles_per_side: number of block-rows (n_blocks);
        defaults to comm.size, the reference's one-rank-per-block layout.
        :param banded: arrow-banded decomposition (blocks (r, r±1) carried;
        ±1 halo X exchange per iteration — the ArrowMPI variant,
        arrow_mpi.py:123-175,211-219)."""
        self.comm = comm if comm is not None else Comm()
        self.banded = banded
        self.column_comm = self.comm
        self.tiles_per_side = tiles_per_side if tiles_per_side is not None else self.comm.size
        assert self.tiles_per_side >= 1
        self.device = device
        self.backend = make_backend(device)

        # contiguous block span owned by this rank
        bpr = -(-self.tiles_per_side // self.comm.size)
        self.blocks_per_rank = bpr
        self.first_block = min(self.comm.rank * bpr, self.tiles_per_side)
        self.last_block = min(self.first_block + bpr, self.tiles_per_side)
        self.n_owned = self.last_block - self.first_block

        self.width: Optional[int] = None
        self.A_0i: List = []          # A_0c for owned c (product handles or scipy)
        self.A_ii: List = []          # A_rr for owned r > 0 (None at r == 0)
        self.A_i0: List = []          # A_r0 for owned r > 0
        self.X_i: Optional[torch.Tensor] = None   # (n_owned*width, k) stripe
        self.X_0: Optional[torch.Tensor] = None   # (width, k)
        self.C_i: Optional[torch.Tensor] = None   # (n_owned*width, k) stripe
        self.C_0: Optional[torch.Tensor] = None   # (width, k) partial/reduced
        self.nnz_owned = 0
        self.A_lo: List = []          # A_{r,r-1} for owned r >= 2 (banded)
        self.A_hi: List = []          # A_{r,r+1} for owned r < nb-1 (banded)
        self.X_halo_lo = None         # X_{first-1} tile from the prev rank
        self.X_halo_hi = None         # X_{last} tile from the next rank
        self._A_bd_lo = None          # boundary (first, first-1) vs halo_lo
        self._A_bd_hi = None          # boundary (last-1, last) vs halo_hi
        self._A_all = None            # single-process fully-fused structure
        self._A_col = None            # A/B: X_0-column entries, hub-sorted
        # iterated-loop optimisation (bench/cfg5 semantics, X := C between
        # steps): replace iteration t's C_0 Reduce + iteration t+1's X_0
        # Bcast with ONE allreduce — next X_0 IS the reduced C_0. Off by
        # default (reference-parity mode sets fresh X each iteration).
        self.allreduce_x0 = False
        self._x0_valid = False
        self._prev_result = None
        # deferred C_0-allreduce pipeline (allreduce_x0 + world>1, slim):
        # the collective posted in step t is waited in step t+1 just before
        # its first consumer, so it overlaps BOTH the rest launch of step t
        # and the row-0 compute of step t+1 (ranks owning block 0 wait
        # first and refresh their stripe head from the reduced X_0 — the
        # only rows whose consumer is the row-0 launch itself)
        self._pending_x0_works: list = []
        # merged resident GPU structures (built at load time, gpu only)
        self._A_row0 = None
        self._A_rest = None
        self._rest_row_offset = 0
        # optional HIP-event instrumentation of the SpMM kernel launches
        # (filled by bench.py for the roofline; list of
        # (start_event, end_event, nnz, c_rows, x_rows) tuples)
        self.kernel_events: Optional[list] = None

    # -- data loading --------------------------------------------------------

    def load_sparse_matrix_from_blocks(self, blocks) -> None:
        """blocks: block grid (list of lists; entries not owned by this rank
        may be None). Owned pieces are uploaded once and stay resident.

        GPU path: the owned blocks are MERGED into two resident structures
         so each iteration is two fused launches
        instead of 3·blocks:
          * `_A_row0`: all owned A_0c side by side (w x n_owned·w), read
            against the X stripe -> C_0 written once;
          * `_A_rest`: per-row merge of A_rr (columns -> stripe) and A_r0
            (columns negative-encoded -> X_0) for owned r > 0 -> C written
            once per row (the reference multiplies and accumulates
            separately, arrow_slim_mpi.py:121-144).
        CPU path keeps per-block scipy (the reference's own cpu dataflow).
        """
        assert len(blocks) == self.tiles_per_side
        nb = self.tiles_per_side
        self.A_0i, self.A_ii, self.A_i0 = [], [], []
        self.A_lo, self.A_hi = [], []
        self.nnz_owned = 0
        gpu = self.backend.device == 'cuda'
        for r in range(self.first_block, self.last_block):
            b0r = blocks[0][r]
            assert b0r is not None, f"missing block (0,{r})"
            if self.width is None:
                self.width = b0r.shape[0]
            self.A_0i.append(b0r if gpu else self.backend.upload_block(b0r))
            self.nnz_owned += b0r.nnz
            if r > 0:
                assert blocks[r][r] is not None and blocks[r][0] is not None, \
                    f"missing diagonal/column block for row {r}"
                rr, r0 = blocks[r][r], blocks[r][0]
                self.A_ii.append(rr if gpu else self.backend.upload_block(rr))
                self.A_i0.append(r0 if gpu else self.backend.upload_block(r0))
                self.nnz_owned += rr.nnz + r0.nnz
            else:
                self.A_ii.append(None)
                self.A_i0.append(None)
            # banded off-diagonals (arrow_mpi.py:459-469): (r, r-1) for
            # r >= 2, (r, r+1) for 0 < r < nb-1
            lo = blocks[r][r - 1] if (self.banded and r >= 2) else None
            hi = blocks[r][r + 1] if (self.banded and 0 < r < nb - 1) else None
            self.A_lo.append(lo if (gpu or lo is None) else self.backend.upload_block(lo))
            self.A_hi.append(hi if (gpu or hi is None) else self.backend.upload_block(hi))
            self.nnz_owned += (lo.nnz if lo is not None else 0) + \
                              (hi.nnz if hi is not None else 0)
        if gpu:
            self._build_merged_gpu()
            # drop the host block references — device copies are resident
            self.A_0i = self.A_ii = self.A_i0 = None
            self.A_lo = self.A_hi = None
        else:
            self._optimize_Ai_slices()

    def _optimize_Ai_slices(self, threshold: float = 0.3) -> None:
        """Column-compact cpu blocks with < threshold nonzero columns,
        keeping the nnz-column map used to gather X at multiply time
        (reference arrow_slim_mpi.py:329-350). On the GPU this is moot: the
        resident merged structure touches only the X rows its indices name,
        so there is nothing to compact."""
        self._nnz_columns = [[None, None, None] for _ in range(max(self.n_owned, 1))]
        for j in range(self.n_owned):
            for slot, blocks in ((0, self.A_0i), (1, self.A_ii), (2, self.A_i0)):
                b = blocks[j]
                if b is None:
                    continue
                nnz_cols = np.unique(b.nonzero()[1])
                if len(nnz_cols) < threshold * b.shape[1]:
                    blocks[j] = b[:, nnz_cols]
                    self._nnz_columns[j][slot] = nnz_cols

    def _row0_chunks(self, w: int) -> int:
        """Chunk count of the C_0 (all)reduce pipeline — a pure function of
        (width, world size) so every rank issues the identical collective
        schedule, including ranks that own no blocks of this matrix."""
        import os as _os
        n_chunks = 4 if (w >= 64 and self.comm.size > 1) else 1
        env_c = _os.environ.get('ARROW_ROW0_CHUNKS')
        if env_c:
            n_chunks = max(1, min(int(env_c), w))
        return n_chunks

    @property
    def _row0_col_items(self) -> int:
        """Work-item ordering of the row-0 (hub) structures (A/B knob; see
        arrow_csr_create_opts in include/arrow_spmm.h): 0 row order,
        1 global column sort (a queue segment becomes a column window of
        X), 2 column sort WITHIN each nnz-balanced row segment (two-level:
        each XCD keeps its contiguous C_0 range but walks its X gathers in
        column order)."""
        import os as _os
        v = _os.environ.get('ARROW_ROW0_COLSORT', '0')
        return int(v) if v in ('0', '1', '2') else 0

    def _build_merged_gpu(self) -> None:
        w = self.width
        nw = self.n_owned
        self._A_row0 = None
        self._A_rest = []
        self._A_all = None
        self._rest_row_offset = 0
        if nw == 0:
            return
        # DEFAULT ON at world=1 since round 2: under the per-XCD queue
        # scheduler the single fused launch reads the X stripe ONCE per
        # iteration (the round-1 "-8%" verdict predates the queue
        # scheduler): measured +12% at 100M (4435 vs 3966 GF/s) and +22%
        # at 20M (gpurun_out/r02_ab.log). ARROW_FUSE_ALL=0 restores the
        # two-launch layout (required by ARROW_SPLIT_COL/ARROW_PAR_ROW0).
        import os as _os
        fuse_all = (_os.environ.get('ARROW_FUSE_ALL', '1') != '0'
                    and self.comm.size == 1 and self.first_block == 0
                    and self.n_owned == self.tiles_per_side)
        # --- row-0 merge: C_0 = [A_0,first .. A_0,last-1] @ X_stripe -------
        # Built in ROW CHUNKS so each chunk's C_0 slice can start its
        # (all)reduce while later chunks and the rest launch still compute:
        # at N=8 the 6.4 GB collective pipelines behind compute instead of
        # serialising after one monolithic launch.
        rows_cat, cols_cat, data_cat = [], [], []
        for j, b in enumerate(self.A_0i):
            b = b.tocsr()
            rows_cat.append(np.repeat(np.arange(w), np.diff(b.indptr)))
            cols_cat.append(b.indices.astype(np.int64) + j * w)
            data_cat.append(b.data.astype(np.float32))
        if fuse_all:
            # single process: block-row 0's reduce is the identity, so the
            # whole matrix fuses into ONE launch writing C in place — the
            # X stripe is read once per iteration instead of twice, and
            # X_0 == X_i[:w] needs no broadcast copy
            row0_sets = (rows_cat, cols_cat, data_cat)
        else:
            n_chunks = self._row0_chunks(w)
            rows = np.concatenate(rows_cat)
            cols = np.concatenate(cols_cat)
            data = np.concatenate(data_cat)
            bounds = [w * q // n_chunks for q in range(n_chunks + 1)]
            self._A_row0 = []
            qb_row0 = int(_os.environ.get('ARROW_Q_BLOCKS_ROW0', '0'))
            for q in range(n_chunks):
                lo, hi = bounds[q], bounds[q + 1]
                m = (rows >= lo) & (rows < hi)
                h = self._merged_handle(hi - lo, nw * w,
                                        [rows[m] - lo], [cols[m]], [data[m]],
                                        x_rows=nw * w // n_chunks,
                                        col_items=self._row0_col_items)
                if qb_row0:
                    h.set_qblocks(qb_row0)
                self._A_row0.append((h, lo, hi))
        # --- rest merge: C[r] = A_rr @ X_r + A_r0 @ X_0
        #     (+ interior banded off-diagonals A_{r,r±1} @ X_{r±1} when the
        #      neighbour block is owned; boundary off-diagonals become small
        #      separate structures against the halo tiles) ----------------
        self._rest_row_offset = w if self.first_block == 0 else 0
        self._A_bd_lo = self._A_bd_hi = None
        lo_list = self.A_lo if self.A_lo else [None] * nw
        hi_list = self.A_hi if self.A_hi else [None] * nw
        if fuse_all:
            self._rest_row_offset = 0
            rows_cat, cols_cat, data_cat = row0_sets
        else:
            rows_cat, cols_cat, data_cat = [], [], []
        # A/B (ARROW_SPLIT_COL=1): keep the A_r0 (X_0) entries OUT of the
        # rest structure and process them in a separate launch whose row
        # order is sorted by hub column — the zipf-skewed X_0 gather then
        # walks the hot head near-sequentially (cache-friendly) at the cost
        # of one extra C read-modify-write pass
        split_col = (_os.environ.get('ARROW_SPLIT_COL', '0') == '1'
                     and not fuse_all)
        col_rows, col_cols, col_data = [], [], []
        n_rest = 0
        for j, r in enumerate(range(self.first_block, self.last_block)):
            if r == 0:
                continue
            rr = self.A_ii[j].tocsr()
            r0 = self.A_i0[j].tocsr()
            local = j * w - self._rest_row_offset
            rows_cat.append(local + np.repeat(np.arange(w), np.diff(rr.indptr)))
            cols_cat.append(rr.indices.astype(np.int64) + j * w)
            data_cat.append(rr.data.astype(np.float32))
            if split_col:
                col_rows.append(local + np.repeat(np.arange(w), np.diff(r0.indptr)))
                col_cols.append(-(r0.indices.astype(np.int64) + 1))  # -> X_0
                col_data.append(r0.data.astype(np.float32))
            else:
                rows_cat.append(local + np.repeat(np.arange(w), np.diff(r0.indptr)))
                cols_cat.append(-(r0.indices.astype(np.int64) + 1))  # -> X_0
                data_cat.append(r0.data.astype(np.float32))
            for off, blk in ((-1, lo_list[j]), (1, hi_list[j])):
                if blk is None:
                    continue
                t = r + off
                if self.first_block <= t < self.last_block:  # interior
                    bb = blk.tocsr()
                    rows_cat.append(local + np.repeat(np.arange(w), np.diff(bb.indptr)))
                    cols_cat.append(bb.indices.astype(np.int64) + (t - self.first_block) * w)
                    data_cat.append(bb.data.astype(np.float32))
                elif off < 0:  # boundary vs X_halo_lo
                    self._A_bd_lo = (self.backend.upload_block(blk), local)
                else:          # boundary vs X_halo_hi
                    self._A_bd_hi = (self.backend.upload_block(blk), local)
            n_rest += 1
        if fuse_all:
            # x_rows = stripe only: the X_0 operand aliases the stripe head,
            # so its rows are not extra algorithmic traffic
            self._A_all = self._merged_handle(nw * w, nw * w, rows_cat,
                                              cols_cat, data_cat,
                                              x_rows=nw * w)
        elif n_rest:
            rest_rows = nw * w - self._rest_row_offset
            x_rows_total = (n_rest * w + w if not split_col else n_rest * w)
            # Auto-split into row-range chunks when the merged nonzero count
            # would overflow the per-structure int32 limit (scale headroom
            # beyond cfg4); ARROW_REST_CHUNK_NNZ forces small chunks in tests.
            cap = int(_os.environ.get('ARROW_REST_CHUNK_NNZ', 1 << 30))
            self._A_rest = self._merged_chunks(rest_rows, nw * w, rows_cat,
                                               cols_cat, data_cat,
                                               x_rows_total, cap)
            qb_rest = int(_os.environ.get('ARROW_Q_BLOCKS_REST', '0'))
            if qb_rest:
                for h, _, _ in self._A_rest:
                    h.set_qblocks(qb_rest)
            if split_col and col_rows:
                self._A_col = self._merged_col_sorted(rest_rows, nw * w,
                                                      col_rows, col_cols,
                                                      col_data)

    def _merged_chunks(self, n_rows, n_cols, rows_cat, cols_cat, data_cat,
                       x_rows_total, cap):
        """One merged structure, or several row-range chunks when the
        nonzero count would overflow the per-structure int32 limit."""
        rows = np.concatenate(rows_cat)
        total_nnz = rows.size
        if total_nnz <= cap:
            h = self._merged_handle(n_rows, n_cols, [rows], cols_cat,
                                    data_cat, x_rows=x_rows_total)
            return [(h, 0, n_rows)]
        cols = np.concatenate(cols_cat)
        data = np.concatenate(data_cat)
        per_row = np.bincount(rows, minlength=n_rows)
        cum = np.cumsum(per_row)
        n_chunks = int(np.ceil(total_nnz / cap))
        out = []
        lo = 0
        for q in range(n_chunks):
            target = total_nnz * (q + 1) // n_chunks
            hi = int(np.searchsorted(cum, target)) + 1 if q < n_chunks - 1 else n_rows
            hi = max(min(hi, n_rows), lo + 1)
            m = (rows >= lo) & (rows < hi)
            h = self._merged_handle(hi - lo, n_cols, [rows[m] - lo],
                                    [cols[m]], [data[m]],
                                    x_rows=max(1, x_rows_total // n_chunks))
            out.append((h, lo, hi))
            lo = hi
            if lo >= n_rows:
                break
        return out

    def _merged_col_sorted(self, n_rows, n_cols, rows_cat, cols_cat, data_cat):
        """Build the X_0-entry structure with ROW BLOCKS ordered by their
        hottest (smallest) hub column; each row appears once (exclusive C
        writes via explicit row ids), so the launch is a plain beta=1 pass."""
        rows = np.concatenate(rows_cat)
        cols = np.concatenate(cols_cat)
        data = np.concatenate(data_cat)
        x0col = -cols - 1
        order = np.lexsort((x0col, rows))  # group by row, min col first
        rows, cols, data, x0col = rows[order], cols[order], data[order], x0col[order]
        uniq_rows, starts = np.unique(rows, return_index=True)
        key = x0col[starts]                      # per-row min hub column
        row_rank = np.argsort(key, kind='stable')
        perm_rows = uniq_rows[row_rank]          # output rows in key order
        remap = np.full(n_rows, -1, dtype=np.int64)
        remap[perm_rows] = np.arange(perm_rows.size)
        new_rows = remap[rows]
        order2 = np.argsort(new_rows, kind='stable')
        cols2, data2 = cols[order2], data[order2]
        counts = np.bincount(new_rows, minlength=perm_rows.size)
        indptr = np.zeros(perm_rows.size + 1, dtype=np.int64)
        np.cumsum(counts, out=indptr[1:])
        handle = self.backend.upload_arrays(
            (perm_rows.size, n_cols), indptr, cols2.astype(np.int32),
            data2.astype(np.float32), row_ids=perm_rows.astype(np.int64))
        handle.x_rows = self.width
        return handle

    def _merged_handle(self, n_rows, n_cols, rows_cat, cols_cat, data_cat,
                       x_rows, col_items=False):
        rows = np.concatenate(rows_cat)
        cols = np.concatenate(cols_cat)
        data = np.concatenate(data_cat)
        order = np.argsort(rows, kind='stable')
        rows, cols, data = rows[order], cols[order], data[order]
        indptr = np.zeros(n_rows + 1, dtype=np.int64)
        np.cumsum(np.bincount(rows, minlength=n_rows), out=indptr[1:])
        handle = self.backend.upload_arrays((n_rows, n_cols), indptr,
                                            cols.astype(np.int32),
                                            data.astype(np.float32),
                                            col_items=col_items)
        handle.x_rows = x_rows  # algorithmic X rows for the roofline
        return handle

    # -- buffers -------------------------------------------------------------

    def zero_rhs(self, number_of_rows_per_rank: int, number_of_columns: int,
                 dtype=np.float32) -> None:
        assert number_of_rows_per_rank >= 1 and number_of_columns >= 1
        if np.dtype(dtype) != np.float32:
            # the reference chooses the dtype at zero_rhs time
            # (arrow_slim_mpi.py:354)
            if self.backend.device == 'cuda':
                raise NotImplementedError(
                    "float64 runs on device='cpu' (HIP kernels are fp32)")
            if np.dtype(dtype) != self.backend.np_dtype:
                self.backend = make_backend(self.device, dtype)
                self.C_i = self.C_0 = self.X_i = self.X_0 = None
        if self.width is not None and self.nnz_owned:
            # blocks were loaded at a fixed block size; a different width
            # would silently mis-tile the resident structures against the
            # X/C buffers — refuse instead
            assert number_of_rows_per_rank == self.width, (
                f"zero_rhs width {number_of_rows_per_rank} != block size "
                f"{self.width} fixed at load time")
        if self.backend.device == 'cuda':
            # loud capacity check BEFORE allocating: the buffer set is
            # 2 stripes + 2 head tiles (+2 halos banded) of fp32; a k that
            # busts 288 GB of HBM must fail with a sizing message, not a
            # mid-iteration OOM (VERDICT r1 "missing" #4; the reference's
            # own k-tiling, spmm_petsc.py:323-395, is moot at this
            # capacity but the guard is not). Only bytes that would be
            # NEWLY allocated count (a repeated zero_rhs with matching
            # shapes reuses the buffers).
            import torch as _t
            w_, k_ = number_of_rows_per_rank, number_of_columns
            stripe_sh = (max(self.n_owned, 1) * w_, k_)
            shapes = [('C_i', stripe_sh), ('X_i', stripe_sh),
                      ('C_0', (w_, k_)), ('X_0', (w_, k_))]
            if self.banded:
                shapes += [('X_halo_lo', (w_, k_)), ('X_halo_hi', (w_, k_))]
            need = sum(4 * sh[0] * sh[1] for name, sh in shapes
                       if getattr(self, name) is None
                       or tuple(getattr(self, name).shape) != sh)
            if need:
                free, total = _t.cuda.mem_get_info()
                if need > free:
                    raise MemoryError(
                        f"zero_rhs: feature/result buffers need "
                        f"{need/2**30:.1f} GiB but only {free/2**30:.1f} "
                        f"GiB of HBM are free (k={k_}, width={w_}, "
                        f"{self.n_owned} owned blocks). Shard over more "
                        f"GPUs or reduce k.")
        self.width = number_of_rows_per_rank
        w, k = number_of_rows_per_rank, number_of_columns
        stripe = (max(self.n_owned, 1) * w, k)
        for name, shape in (('C_i', stripe), ('C_0', (w, k)),
                            ('X_i', stripe), ('X_0', (w, k))):
            buf = getattr(self, name)
            if buf is None or tuple(buf.shape) != shape:
                setattr(self, name, self.backend.zeros(shape))
            else:
                buf.zero_()
        if self.banded:
            for name in ('X_halo_lo', 'X_halo_hi'):
                buf = getattr(self, name)
                if buf is None or tuple(buf.shape) != (w, k):
                    setattr(self, name, self.backend.zeros((w, k)))
                else:
                    buf.zero_()
        self._x0_valid = False  # buffers were just reset
        # ping-pong pool: spmm writes C into a stripe that does NOT alias
        # X_i, so `X := C` between iterations (set_features(result_tile()))
        # is race-free on the GPU (the reference allocates a fresh C every
        # iteration instead, arrow_slim_mpi.py:125-127)
        self._stripe_bufs = [self.C_i, self.X_i]

    def _select_result_buffer(self):
        if self.X_i is None or self.C_i.data_ptr() != self.X_i.data_ptr():
            return
        for buf in self._stripe_bufs:
            if buf.data_ptr() != self.X_i.data_ptr():
                self.C_i = buf
                return
        self.C_i = self.backend.zeros(tuple(self.X_i.shape))
        self._stripe_bufs.append(self.C_i)

    def set_features(self, X) -> None:
        """Stores a reference (no copy) when X already lives on this
        engine's device; otherwise moves it there once."""
        assert X is not None
        if isinstance(X, torch.Tensor) and (X.device.type == self.backend.device
                                            or (self.backend.device == 'cuda' and X.is_cuda)):
            self.X_i = X
        else:
            self.X_i = self.backend.asarray(X)

    def feature_tile(self):
        return self.X_i

    def set_features_slice_from_features(self, X) -> None:
        """Deprecated in the reference (arrow_slim_mpi.py:434-440) but part
        of the kept ABC: set this rank's stripe from the FULL feature
        matrix."""
        w = self.width
        assert X.shape[0] % self.tiles_per_side == 0 or \
            X.shape[0] >= self.last_block * w
        self.set_features(X[self.first_block * w: self.last_block * w])

    def result_tile(self):
        return self.C_i

    def is_column_rank(self) -> bool:
        return True

    # -- compute -------------------------------------------------------------

    def spmm(self, device: str = None) -> None:
        """One arrow SpMM (reference _arrow_spmm, arrow_slim_mpi.py:246-280).
        `device` is accepted for API compatibility; the engine's device was
        fixed at construction (no silent fallback)."""
        if device is not None and device != self.device:
            raise ValueError(
                f"engine was built for device={self.device!r}; got {device!r}")
        self._select_result_buffer()
        w = self.width

        # X_0 broadcast (arrow_slim_mpi.py:265-273); owner of block 0 is
        # rank 0. Overlapped: the row-0 launch below reads only the stripe,
        # so the broadcast runs concurrently with it (the reference's
        # broadcast is blocking, arrow_slim_mpi.py:273).
        tic = time.perf_counter()
        bcast_work = None
        x0_ready = (self.allreduce_x0 and self._x0_valid
                    and self.X_i is self._prev_result)
        if self._pending_x0_works and not x0_ready:
            # features were replaced outside the iterated loop while a
            # deferred allreduce is still in flight on X_0 — drain it
            # before the buffer is overwritten below
            for wk in self._pending_x0_works:
                wk.wait()
            self._pending_x0_works = []
        if self._A_all is None and not x0_ready:
            if self.first_block == 0 and self.n_owned > 0:
                self.X_0.copy_(self.X_i[:w])
            bcast_work = self.comm.bcast_(self.X_0, src=0, async_op=True)
        wb_logging.log({"spmm_x_bcast_time": time.perf_counter() - tic})

        if self.banded:
            self._exchange_halos()

        tic = time.perf_counter()
        if self.backend.device == 'cuda':
            self._spmm_gpu(bcast_work)
        else:
            if bcast_work is not None:
                bcast_work.wait()
            self._spmm_cpu()
        wb_logging.log({"spmm_kernel_time": time.perf_counter() - tic})

        # the reduced C_0 is block-row 0's result (arrow_slim_mpi.py:152-155);
        # in the fully-fused single-process layout it was written in place.
        # In the DEFERRED allreduce pipeline the head copy moves to the
        # start of the next step (_spmm_gpu refreshes the stripe head from
        # the reduced X_0 before the row-0 launch — the head's only
        # consumer), so the iterated loop's values are unchanged while the
        # collective overlaps this step's rest launch AND the next step's
        # row-0 compute.
        if (self._A_all is None and self.first_block == 0
                and self.n_owned > 0 and self.comm.rank == 0
                and not self._pending_x0_works):
            self.C_i[:w].copy_(self.C_0)
        if self.allreduce_x0 and self._A_all is None \
                and self.backend.device == 'cuda':
            # the allreduced C_0 becomes the next iteration's X_0
            self.X_0, self.C_0 = self.C_0, self.X_0
            self._x0_valid = True
            self._prev_result = self.C_i

    def _timed(self, fn, nnz, c_rows, x_rows):
        """HIP-event instrumentation for the roofline (bench.py)."""
        if self.kernel_events is None:
            fn()
            return
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        fn()
        e.record()
        self.kernel_events.append((s, e, nnz, c_rows, x_rows))

    def _spmm_gpu(self, bcast_work=None) -> None:
        """Two fused launches on the resident merged structures
        (DESIGN.md §kernels), with the X_0 broadcast and the C_0 reduce
        overlapped with compute (RCCL runs on its own stream; `wait()`
        inserts stream dependencies, not host blocks)."""
        be = self.backend
        w = self.width

        # deferred-allreduce pipeline: drain LAST step's C_0 collective at
        # its first consumer. Ranks owning block 0 consume the head rows in
        # their row-0 launch — wait now and refresh the stripe head from
        # the reduced X_0 (the deferred equivalent of last step's
        # C_i[:w].copy_(C_0)); other ranks wait later, just before the
        # rest launch (their only X_0 consumer) so the collective also
        # overlaps this step's row-0 compute.
        pend = self._pending_x0_works
        if pend and self.first_block == 0 and self.n_owned > 0:
            for wk in pend:
                wk.wait()
            self.X_i[:w].copy_(self.X_0)
            pend = []
            self._pending_x0_works = []

        if self._A_all is not None:
            # single process: whole matrix in ONE fused launch, C in place;
            # X_0 == X_i[:w] so the second operand is the stripe itself
            h = self._A_all
            self._timed(lambda: be.spmm_dual(h, self.X_i, self.X_i, self.C_i, 0),
                        h.nnz, self.n_owned * w, h.x_rows)
            return

        # C_0 = A_row0_merged @ X_stripe in row chunks (the reference runs
        # one CSRMM per block and re-uploads A and X,
        # arrow_slim_mpi.py:181-195); each chunk's (all)reduce starts as
        # soon as its slice is computed and pipelines behind the remaining
        # compute (arrow_slim_mpi.py:116's single blocking Reduce).
        #
        # ARROW_PAR_ROW0=1: launch the row-0 (hub) structure on a SIDE HIP
        # stream so it runs CONCURRENTLY with the rest launch below — the
        # hub gathers are fabric-bound while the banded rest launch is not,
        # so co-running them fills the slack. Fork-join via
        # events (capture-legal for hipGraph).
        import os as _os
        par = (_os.environ.get('ARROW_PAR_ROW0', '0') == '1'
               and bool(self._A_row0))
        stream_ctx = None
        if par:
            if not hasattr(self, '_row0_stream') or self._row0_stream is None:
                self._row0_stream = torch.cuda.Stream()
            ev_x = torch.cuda.Event()
            ev_x.record()  # X_i (and X_0 copy) ready on the main stream
            self._row0_stream.wait_event(ev_x)
            stream_ctx = torch.cuda.stream(self._row0_stream)
            stream_ctx.__enter__()
        tic = time.perf_counter()
        reduce_works = []
        if self._A_row0:
            for h, lo, hi in self._A_row0:
                C_sl = self.C_0[lo:hi]
                self._timed(lambda h=h, C_sl=C_sl:
                            be.spmm_block(h, self.X_i, C_sl, 0),
                            h.nnz, hi - lo, h.x_rows)
                if self.allreduce_x0:
                    reduce_works.append(
                        self.comm.allreduce_sum_(C_sl, async_op=True))
                else:
                    reduce_works.append(
                        self.comm.reduce_sum_(C_sl, dst=0, async_op=True))
        else:
            # a rank with no blocks of this matrix must still issue the SAME
            # collective schedule as its peers (chunk count is a function of
            # (width, world) only) — a single reduce here would deadlock
            # against their chunked reduces
            self.C_0.zero_()
            n_chunks = self._row0_chunks(w)
            bounds = [w * q // n_chunks for q in range(n_chunks + 1)]
            for q in range(n_chunks):
                C_sl = self.C_0[bounds[q]:bounds[q + 1]]
                if self.allreduce_x0:
                    reduce_works.append(
                        self.comm.allreduce_sum_(C_sl, async_op=True))
                else:
                    reduce_works.append(
                        self.comm.reduce_sum_(C_sl, dst=0, async_op=True))
        if par:
            ev_r = torch.cuda.Event()
            ev_r.record(self._row0_stream)
            stream_ctx.__exit__(None, None, None)
        wb_logging.log({"spmm_row_reduce": time.perf_counter() - tic})

        # C_rest = A_diag_merged @ X_stripe + A_col_merged @ X_0 fused:
        # C written once (vs the reference's multiply-then-accumulate,
        # arrow_slim_mpi.py:121-144)
        if pend:
            # ranks without block 0: last step's deferred allreduce must
            # land before the rest launch reads X_0
            for wk in pend:
                wk.wait()
            self._pending_x0_works = []
        if bcast_work is not None:
            bcast_work.wait()  # rest reads X_0
        if self._A_rest:
            for h, lo, hi in self._A_rest:
                C_sub = self.C_i[self._rest_row_offset + lo:
                                 self._rest_row_offset + hi]
                self._timed(lambda h=h, C_sub=C_sub:
                            be.spmm_dual(h, self.X_i, self.X_0, C_sub, 0),
                            h.nnz, C_sub.shape[0], h.x_rows)
        if self._A_col is not None:
            h = self._A_col
            C_full = self.C_i[self._rest_row_offset:self.n_owned * w]
            self._timed(lambda: be.spmm_dual(h, self.X_i, self.X_0, C_full, 1),
                        h.nnz, h.shape[0], h.x_rows)

        # banded boundary off-diagonals against the received halo tiles
        for bd, halo in ((self._A_bd_lo, self.X_halo_lo),
                         (self._A_bd_hi, self.X_halo_hi)):
            if bd is not None:
                hdl, local = bd
                Cr = self.C_i[local:local + w]
                self._timed(lambda: be.spmm_block(hdl, halo, Cr, 1),
                            hdl.nnz, w, w)
        if par:
            # join: the main stream's C_0 consumers (head copy / X_0 swap)
            # wait for the side-stream row-0 work
            torch.cuda.current_stream().wait_event(ev_r)
        import os as _os2
        defer = (self.allreduce_x0 and self.comm.size > 1
                 and not self.banded
                 and _os2.environ.get('ARROW_X0_DEFER', '1') != '0')
        if defer:
            self._pending_x0_works = [wk for wk in reduce_works
                                      if wk is not None]
        else:
            for wk in reduce_works:
                if wk is not None:
                    wk.wait()

    def _exchange_halos(self) -> None:
        """±1 halo X exchange for the banded mode: my first owned tile goes
        to the previous rank (their halo_hi), my last to the next rank
        (their halo_lo) — replaces the reference's Isend/Irecv halo pattern
        (arrow_mpi.py:123-175) with grouped p2p over xGMI."""
        if self.comm.size == 1 or self.n_owned == 0:
            return
        import torch.distributed as dist
        if (self.X_i.is_cuda
                and dist.get_backend(getattr(self.comm, 'group', None)) == 'gloo'):
            # gloo has no CUDA p2p: stage halos through host (validation
            # rigs only; the product multi-GPU path is RCCL)
            return self._exchange_halos_staged()
        w = self.width
        P = self.comm.size
        bpr = self.blocks_per_rank
        prev_rank = (self.first_block - 1) // bpr if self.first_block >= 1 else -1
        next_owner = self.last_block // bpr if self.last_block < self.tiles_per_side else -1
        ops = []
        group = getattr(self.comm, 'group', None)
        if prev_rank >= 0:
            ops.append(dist.P2POp(dist.isend, self.X_i[:w].contiguous(),
                                  prev_rank, group=group))
            ops.append(dist.P2POp(dist.irecv, self.X_halo_lo, prev_rank,
                                  group=group))
        if next_owner >= 0 and next_owner < P:
            ops.append(dist.P2POp(dist.isend,
                                  self.X_i[(self.n_owned - 1) * w:self.n_owned * w].contiguous(),
                                  next_owner, group=group))
            ops.append(dist.P2POp(dist.irecv, self.X_halo_hi, next_owner,
                                  group=group))
        if ops:
            for wk in dist.batch_isend_irecv(ops):
                wk.wait()

    def _exchange_halos_staged(self) -> None:
        """Halo exchange with host staging (gloo + CUDA tensors)."""
        import torch.distributed as dist
        w = self.width
        P = self.comm.size
        bpr = self.blocks_per_rank
        prev_rank = (self.first_block - 1) // bpr if self.first_block >= 1 else -1
        next_owner = self.last_block // bpr if self.last_block < self.tiles_per_side else -1
        group = getattr(self.comm, 'group', None)
        ops = []
        lo_cpu = self.X_halo_lo.cpu() if prev_rank >= 0 else None
        hi_cpu = self.X_halo_hi.cpu() if (0 <= next_owner < P) else None
        if prev_rank >= 0:
            ops.append(dist.P2POp(dist.isend, self.X_i[:w].cpu(), prev_rank,
                                  group=group))
            ops.append(dist.P2POp(dist.irecv, lo_cpu, prev_rank, group=group))
        if 0 <= next_owner < P:
            ops.append(dist.P2POp(
                dist.isend,
                self.X_i[(self.n_owned - 1) * w:self.n_owned * w].cpu(),
                next_owner, group=group))
            ops.append(dist.P2POp(dist.irecv, hi_cpu, next_owner, group=group))
        if ops:
            for wk in dist.batch_isend_irecv(ops):
                wk.wait()
        if lo_cpu is not None:
            self.X_halo_lo.copy_(lo_cpu)
        if hi_cpu is not None:
            self.X_halo_hi.copy_(hi_cpu)

    def _spmm_cpu(self) -> None:
        """Per-block scipy dataflow — the reference's own cpu path
        (arrow_slim_mpi.py:78-156)."""
        be = self.backend
        w = self.width

        def xsel(X, j, slot):
            cols = self._nnz_columns[j][slot] if hasattr(self, '_nnz_columns') else None
            return X.contiguous() if cols is None else X[cols].contiguous()

        first = True
        for j, r in enumerate(range(self.first_block, self.last_block)):
            Xr = self.X_i[j * w:(j + 1) * w]
            be.spmm_block(self.A_0i[j], xsel(Xr, j, 0), self.C_0, 0 if first else 1)
            first = False
        if first:  # rank owns no blocks of this matrix
            self.C_0.zero_()

        self.comm.reduce_sum_(self.C_0, dst=0)

        for j, r in enumerate(range(self.first_block, self.last_block)):
            if r == 0:
                continue
            Xr = self.X_i[j * w:(j + 1) * w]
            Cr = self.C_i[j * w:(j + 1) * w]
            be.spmm_block(self.A_ii[j], xsel(Xr, j, 1), Cr, 0)
            be.spmm_block(self.A_i0[j], xsel(self.X_0, j, 2), Cr, 1)
            # banded ±1 halo terms (arrow_mpi.py:211-219)
            lo = self.A_lo[j] if self.A_lo else None
            hi = self.A_hi[j] if self.A_hi else None
            for off, blk in ((-1, lo), (1, hi)):
                if blk is None:
                    continue
                t = r + off
                if self.first_block <= t < self.last_block:
                    Xn = self.X_i[(t - self.first_block) * w:
                                  (t - self.first_block + 1) * w].contiguous()
                else:
                    Xn = self.X_halo_lo if off < 0 else self.X_halo_hi
                be.spmm_block(blk, Xn, Cr, 1)

    # -- result --------------------------------------------------------------

    def _flush_x0_pipeline(self) -> None:
        """Drain the deferred C_0 allreduce (allreduce_x0 world>1 iterated
        loop) and materialise block-row 0's result head — the deferred
        equivalent of the skipped C_i[:w].copy_(C_0). Called before any
        result readout."""
        if not self._pending_x0_works:
            return
        for wk in self._pending_x0_works:
            wk.wait()
        self._pending_x0_works = []
        if self.first_block == 0 and self.n_owned > 0:
            # after the swap X_0 holds the reduced C_0 of the last step
            self.C_i[:self.width].copy_(self.X_0)

    def allgather_result(self, C=None):
        """All-gathers the per-rank result stripes into the full
        (tiles_per_side*width, k) matrix (reference arrow_slim_mpi.py:415-425).
        Returns numpy; fills C in place if given."""
        self._flush_x0_pipeline()
        w = self.width
        k = self.C_i.shape[1]
        pad_rows = self.blocks_per_rank * w
        stripe = self.backend.zeros((pad_rows, k))
        if self.n_owned:
            stripe[:self.n_owned * w].copy_(self.C_i[:self.n_owned * w])
        full = self.comm.allgather_cat(stripe)
        out = full[:self.tiles_per_side * w]
        out_np = out.cpu().numpy()
        if C is not None:
            C[:] = out_np
            return C
        return out_np

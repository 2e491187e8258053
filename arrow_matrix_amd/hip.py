"""ctypes binding to libarrowspmm.so (the C-ABI HIP compute library).

This is the product GPU path. It FAILS LOUDLY if the extension is missing or
a call errors — there is no CPU fallback here (the explicit `--device cpu`
mode of the package is a separate, deliberate scipy path mirroring the
reference's own cpu mode, arrow_slim_mpi.py:78-156).
"""
import ctypes
import os
from typing import Optional

import numpy as np

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), 'libarrowspmm.so')
_lib: Optional[ctypes.CDLL] = None


class ArrowSpmmError(RuntimeError):
    pass


def _load():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise ArrowSpmmError(
            f"libarrowspmm.so not found at {_LIB_PATH}. Build it with "
            f"`make -C arrow_matrix_amd/csrc` (or __graft_entry__.build()). "
            f"The GPU path does not fall back to CPU.")
    lib = ctypes.CDLL(_LIB_PATH)
    lib.arrow_abi_version.restype = ctypes.c_int
    lib.arrow_last_error.restype = ctypes.c_char_p
    lib.arrow_device_count.restype = ctypes.c_int
    lib.arrow_set_device.argtypes = [ctypes.c_int]
    lib.arrow_set_device.restype = ctypes.c_int
    lib.arrow_synchronize.restype = ctypes.c_int
    lib.arrow_csr_create.restype = ctypes.c_int64
    lib.arrow_csr_create.argtypes = [
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32),
        ctypes.POINTER(ctypes.c_float)]
    lib.arrow_csr_create_rows.restype = ctypes.c_int64
    lib.arrow_csr_create_rows.argtypes = [
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32),
        ctypes.POINTER(ctypes.c_float), ctypes.POINTER(ctypes.c_int64)]
    lib.arrow_csr_create_opts.restype = ctypes.c_int64
    lib.arrow_csr_create_opts.argtypes = [
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32),
        ctypes.POINTER(ctypes.c_float), ctypes.POINTER(ctypes.c_int64),
        ctypes.c_int]
    lib.arrow_csr_destroy.argtypes = [ctypes.c_int64]
    lib.arrow_csr_destroy.restype = ctypes.c_int
    lib.arrow_csr_nnz.argtypes = [ctypes.c_int64]
    lib.arrow_csr_nnz.restype = ctypes.c_int64
    lib.arrow_csr_set_xcd_remap.argtypes = [ctypes.c_int64, ctypes.c_int]
    lib.arrow_csr_set_xcd_remap.restype = ctypes.c_int
    lib.arrow_csr_set_queue.argtypes = [ctypes.c_int64, ctypes.c_int]
    lib.arrow_csr_set_queue.restype = ctypes.c_int
    lib.arrow_csr_set_qblocks.argtypes = [ctypes.c_int64, ctypes.c_int]
    lib.arrow_csr_set_qblocks.restype = ctypes.c_int
    lib.arrow_spmm.argtypes = [ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                               ctypes.c_int64, ctypes.c_int, ctypes.c_void_p]
    lib.arrow_spmm.restype = ctypes.c_int
    lib.arrow_spmm_dual.argtypes = [ctypes.c_int64, ctypes.c_void_p,
                                    ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_int64, ctypes.c_int, ctypes.c_void_p]
    lib.arrow_spmm_dual.restype = ctypes.c_int
    for name in ('arrow_gather_rows_f32', 'arrow_scatter_rows_f32',
                 'arrow_scatter_add_rows_f32'):
        fn = getattr(lib, name)
        fn.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                       ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p]
        fn.restype = ctypes.c_int
    for name in ('arrow_permute_rows_f32', 'arrow_permute_add_rows_f32'):
        fn = getattr(lib, name)
        fn.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                       ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                       ctypes.c_void_p]
        fn.restype = ctypes.c_int
    _lib = lib
    return lib


def is_built() -> bool:
    return os.path.exists(_LIB_PATH)


def _check(rc, what: str):
    if rc < 0:
        raise ArrowSpmmError(f"{what} failed: {_load().arrow_last_error().decode()}")
    return rc


class CsrBlockGPU:
    """A CSR block resident in HBM (uploaded ONCE — unlike the reference,
    which re-uploads A every iteration, arrow_slim_mpi.py:184-232)."""

    def __init__(self, csr=None, arrays=None, row_ids=None, col_items=False):
        """csr: scipy CSR, or arrays=(shape, indptr, indices, data) for raw
        uploads (the fused layouts use negative column indices, which scipy
        would reject). row_ids: optional explicit output-row id per
        structure row (reordered layouts). col_items: work-item ordering
        mode (hub structures; see arrow_csr_create_opts flags): 0/False
        row order, 1/True global column sort, 2 column sort WITHIN each
        XCD queue segment (two-level)."""
        lib = _load()
        if arrays is not None:
            (rows, cols), indptr, indices, data = arrays
        else:
            csr = csr.tocsr()
            rows, cols = csr.shape
            indptr, indices, data = csr.indptr, csr.indices, csr.data
        indptr = np.ascontiguousarray(indptr, dtype=np.int64)
        indices = np.ascontiguousarray(indices, dtype=np.int32)
        data = np.ascontiguousarray(data, dtype=np.float32)
        self.shape = (rows, cols)
        self.nnz = int(indices.size)
        if col_items:
            rid = (np.ascontiguousarray(row_ids, dtype=np.int64)
                   if row_ids is not None else None)
            self._handle = _check(lib.arrow_csr_create_opts(
                rows, cols, self.nnz,
                indptr.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
                indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
                data.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                (rid.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))
                 if rid is not None else None), int(col_items)),
                "arrow_csr_create_opts")
        elif row_ids is not None:
            row_ids = np.ascontiguousarray(row_ids, dtype=np.int64)
            self._handle = _check(lib.arrow_csr_create_rows(
                rows, cols, self.nnz,
                indptr.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
                indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
                data.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                row_ids.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))),
                "arrow_csr_create_rows")
        else:
            self._handle = _check(lib.arrow_csr_create(
                rows, cols, self.nnz,
                indptr.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
                indices.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
                data.ctypes.data_as(ctypes.POINTER(ctypes.c_float))),
                "arrow_csr_create")

    def spmm(self, X_ptr: int, C_ptr: int, k: int, beta: int, stream: int = 0):
        """C (+)= A @ X on device pointers (e.g. torch tensor data_ptr())."""
        _check(_load().arrow_spmm(self._handle, X_ptr, C_ptr, k, beta, stream),
               "arrow_spmm")

    def set_xcd_remap(self, enable: bool):
        _check(_load().arrow_csr_set_xcd_remap(self._handle, 1 if enable else 0),
               "arrow_csr_set_xcd_remap")

    def set_queue(self, mode: int):
        """Per-XCD queue scheduler: 1 on, 0 off, -1 follow ARROW_QUEUE env."""
        _check(_load().arrow_csr_set_queue(self._handle, int(mode)),
               "arrow_csr_set_queue")

    def set_qblocks(self, blocks: int):
        """Per-structure queue-grid override (0 = ARROW_Q_BLOCKS default)."""
        _check(_load().arrow_csr_set_qblocks(self._handle, int(blocks)),
               "arrow_csr_set_qblocks")

    def spmm_dual(self, X0_ptr: int, X1_ptr: int, C_ptr: int, k: int,
                  beta: int, stream: int = 0):
        """Fused dual-operand SpMM: negative column indices (encoded at
        upload as -(idx+1)) read X1 (see include/arrow_spmm.h)."""
        _check(_load().arrow_spmm_dual(self._handle, X0_ptr, X1_ptr, C_ptr,
                                       k, beta, stream), "arrow_spmm_dual")

    def __del__(self):
        if getattr(self, '_handle', None) is not None and _lib is not None:
            try:
                _lib.arrow_csr_destroy(self._handle)
            except Exception:
                pass


def set_device(device: int):
    _check(_load().arrow_set_device(device), "arrow_set_device")


def synchronize():
    _check(_load().arrow_synchronize(), "arrow_synchronize")


def gather_rows(src_ptr: int, dst_ptr: int, idx_ptr: int, n: int, k: int, stream: int = 0):
    _check(_load().arrow_gather_rows_f32(src_ptr, dst_ptr, idx_ptr, n, k, stream),
           "arrow_gather_rows_f32")


def scatter_rows(dst_ptr: int, src_ptr: int, idx_ptr: int, n: int, k: int, stream: int = 0):
    _check(_load().arrow_scatter_rows_f32(dst_ptr, src_ptr, idx_ptr, n, k, stream),
           "arrow_scatter_rows_f32")


def scatter_add_rows(dst_ptr: int, src_ptr: int, idx_ptr: int, n: int, k: int, stream: int = 0):
    _check(_load().arrow_scatter_add_rows_f32(dst_ptr, src_ptr, idx_ptr, n, k, stream),
           "arrow_scatter_add_rows_f32")


def permute_rows(dst_ptr, src_ptr, dst_idx_ptr, src_idx_ptr, n, k, stream=0):
    _check(_load().arrow_permute_rows_f32(dst_ptr, src_ptr, dst_idx_ptr,
                                          src_idx_ptr, n, k, stream),
           "arrow_permute_rows_f32")


def permute_add_rows(dst_ptr, src_ptr, dst_idx_ptr, src_idx_ptr, n, k, stream=0):
    _check(_load().arrow_permute_add_rows_f32(dst_ptr, src_ptr, dst_idx_ptr,
                                              src_idx_ptr, n, k, stream),
           "arrow_permute_add_rows_f32")


def abi_version() -> int:
    return _load().arrow_abi_version()

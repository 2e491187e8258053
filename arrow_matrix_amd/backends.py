"""Compute backends for the arrow engine.

- GpuBackend: the product path — torch CUDA tensors for storage/comm,
  hand-written HIP kernels via the C ABI (hip.py / libarrowspmm.so) for all
  compute. FAILS LOUDLY if the extension is missing; never falls back.
- CpuBackend: the reference's own `--device cpu` semantics — scipy CSR `@`
  (arrow_slim_mpi.py:78-156). This is a deliberate feature (it IS the parity
  reference named by BASELINE.json), not a fallback for the GPU path.

Both expose the same small surface so ArrowSlimMPI/ArrowDecompositionMPI
have a single code path.
"""
from typing import Optional

import numpy as np
import torch

from . import hip


_TORCH_DTYPE = {np.dtype(np.float32): torch.float32,
                np.dtype(np.float64): torch.float64}


class CpuBackend:
    device = 'cpu'

    def __init__(self, dtype=np.float32):
        self.np_dtype = np.dtype(dtype)
        self.torch_dtype = _TORCH_DTYPE[self.np_dtype]

    def zeros(self, shape):
        return torch.zeros(shape, dtype=self.torch_dtype)

    def asarray(self, x) -> torch.Tensor:
        if isinstance(x, torch.Tensor):
            return x.to(self.torch_dtype).cpu()
        return torch.from_numpy(np.ascontiguousarray(x, dtype=self.np_dtype))

    def index_tensor(self, idx: np.ndarray) -> torch.Tensor:
        return torch.from_numpy(np.ascontiguousarray(idx, dtype=np.int64))

    def upload_block(self, csr):
        from scipy import sparse
        return sparse.csr_matrix(csr)

    def spmm_block(self, block, X: torch.Tensor, C: torch.Tensor, beta: int):
        """C (+)= block @ X (scipy kernel, the reference CPU path)."""
        res = block @ X.numpy()
        if beta == 0:
            C.numpy()[:] = res
        else:
            C.numpy()[:] += res

    def gather_rows(self, src: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
        return src[idx]

    def scatter_rows(self, dst: torch.Tensor, idx: torch.Tensor, src: torch.Tensor):
        dst[idx] = src

    def scatter_add_rows(self, dst: torch.Tensor, idx: torch.Tensor, src: torch.Tensor):
        dst[idx] += src

    def permute_rows(self, dst, dst_idx, src, src_idx):
        dst[dst_idx] = src[src_idx]

    def permute_add_rows(self, dst, dst_idx, src, src_idx):
        dst[dst_idx] += src[src_idx]

    def synchronize(self):
        pass


class GpuBackend:
    device = 'cuda'

    def __init__(self, device_index: Optional[int] = None):
        if not torch.cuda.is_available():
            raise hip.ArrowSpmmError(
                "GPU backend requested but no HIP device is available")
        if device_index is None:
            device_index = torch.cuda.current_device()
        self.device_index = device_index
        self.torch_device = torch.device('cuda', device_index)
        # Loading the library here makes a missing .so fail at construction.
        hip.set_device(device_index)

    def _stream(self) -> int:
        return torch.cuda.current_stream(self.device_index).cuda_stream

    def zeros(self, shape):
        return torch.zeros(shape, dtype=torch.float32, device=self.torch_device)

    def asarray(self, x) -> torch.Tensor:
        if isinstance(x, torch.Tensor):
            return x.to(self.torch_device, dtype=torch.float32)
        return torch.from_numpy(np.ascontiguousarray(x, dtype=np.float32)).to(self.torch_device)

    def index_tensor(self, idx: np.ndarray) -> torch.Tensor:
        return torch.from_numpy(np.ascontiguousarray(idx, dtype=np.int64)).to(self.torch_device)

    def upload_block(self, csr):
        return hip.CsrBlockGPU(csr)

    def spmm_block(self, block: hip.CsrBlockGPU, X: torch.Tensor, C: torch.Tensor, beta: int):
        assert X.is_contiguous() and C.is_contiguous()
        block.spmm(X.data_ptr(), C.data_ptr(), C.shape[1], beta, self._stream())

    def upload_arrays(self, shape, indptr, indices, data, row_ids=None,
                      col_items=False) -> hip.CsrBlockGPU:
        return hip.CsrBlockGPU(arrays=(shape, indptr, indices, data),
                               row_ids=row_ids, col_items=col_items)

    def spmm_dual(self, block: hip.CsrBlockGPU, X0: torch.Tensor,
                  X1: torch.Tensor, C: torch.Tensor, beta: int):
        assert X0.is_contiguous() and X1.is_contiguous() and C.is_contiguous()
        block.spmm_dual(X0.data_ptr(), X1.data_ptr(), C.data_ptr(),
                        C.shape[1], beta, self._stream())

    def gather_rows(self, src: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
        out = torch.empty((idx.shape[0], src.shape[1]), dtype=torch.float32,
                          device=self.torch_device)
        if idx.shape[0]:
            hip.gather_rows(src.data_ptr(), out.data_ptr(), idx.data_ptr(),
                            idx.shape[0], src.shape[1], self._stream())
        return out

    def scatter_rows(self, dst: torch.Tensor, idx: torch.Tensor, src: torch.Tensor):
        if idx.shape[0]:
            hip.scatter_rows(dst.data_ptr(), src.data_ptr(), idx.data_ptr(),
                             idx.shape[0], dst.shape[1], self._stream())

    def scatter_add_rows(self, dst: torch.Tensor, idx: torch.Tensor, src: torch.Tensor):
        if idx.shape[0]:
            hip.scatter_add_rows(dst.data_ptr(), src.data_ptr(), idx.data_ptr(),
                                 idx.shape[0], dst.shape[1], self._stream())

    def permute_rows(self, dst, dst_idx, src, src_idx):
        if dst_idx.shape[0]:
            hip.permute_rows(dst.data_ptr(), src.data_ptr(), dst_idx.data_ptr(),
                             src_idx.data_ptr(), dst_idx.shape[0], dst.shape[1],
                             self._stream())

    def permute_add_rows(self, dst, dst_idx, src, src_idx):
        if dst_idx.shape[0]:
            hip.permute_add_rows(dst.data_ptr(), src.data_ptr(),
                                 dst_idx.data_ptr(), src_idx.data_ptr(),
                                 dst_idx.shape[0], dst.shape[1], self._stream())

    def synchronize(self):
        torch.cuda.synchronize(self.device_index)


def make_backend(device: str, dtype=np.float32):
    if device in ('gpu', 'cuda'):
        if np.dtype(dtype) != np.float32:
            raise NotImplementedError(
                "the HIP kernels compute in fp32 (the reference benchmark "
                "default, arrow_bench.py:21); float64 runs on device='cpu'")
        return GpuBackend()
    if device == 'cpu':
        return CpuBackend(dtype)
    raise ValueError(f"unknown device {device!r} (use 'cpu' or 'gpu')")

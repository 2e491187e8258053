"""Communication layer: RCCL over xGMI (torch.distributed backend "nccl")
on GPU, gloo on CPU (tests), or a no-op single-process comm.

Replaces the reference's mpi4py collectives (full call-site inventory in
SURVEY.md §2): Bcast -> broadcast, Reduce(SUM) -> reduce, Alltoallv ->
batched isend/irecv (RCCL has no alltoallv primitive; grouped ncclSend/Recv
via torch.distributed.batch_isend_irecv is the native idiom),
Gather+Bcast (allgather_result) -> all_gather, allreduce(LOR) -> all_reduce(MAX).

One process per GPU; all payloads are torch tensors (CUDA tensors under
nccl, CPU tensors under gloo — so the same engine code runs the gloo
world_size>1 CPU tests and the RCCL GPU path).
"""
from typing import List, Sequence

import torch

try:
    import torch.distributed as dist
except Exception:  # pragma: no cover
    dist = None


class Comm:
    """Single-process comm (world size 1): everything is local."""

    rank = 0
    size = 1

    def barrier(self):
        pass

    def bcast_(self, tensor: torch.Tensor, src: int, async_op: bool = False):
        return None

    def reduce_sum_(self, tensor: torch.Tensor, dst: int, async_op: bool = False):
        return None

    def allreduce_max_(self, tensor: torch.Tensor):
        pass

    def allreduce_sum_(self, tensor: torch.Tensor, async_op: bool = False):
        return None

    def alltoallv(self, send: torch.Tensor, send_counts: Sequence[int],
                  recv_counts: Sequence[int]) -> torch.Tensor:
        assert len(send_counts) == 1 and len(recv_counts) == 1
        assert send_counts[0] == recv_counts[0]
        return send

    def alltoallv_async(self, send: torch.Tensor, send_counts, recv_counts):
        assert len(send_counts) == 1 and len(recv_counts) == 1
        return send, []

    def allgather_cat(self, tensor: torch.Tensor) -> torch.Tensor:
        return tensor

    def allgather_int(self, value: int) -> List[int]:
        return [int(value)]

    def alltoall_ints(self, values: Sequence[int]) -> List[int]:
        assert len(values) == 1
        return [int(values[0])]


class TorchDistComm(Comm):
    """torch.distributed-backed comm (backend "nccl" == RCCL on ROCm, or
    "gloo" for CPU tests)."""

    def __init__(self, group=None):
        assert dist is not None and dist.is_initialized(), \
            "torch.distributed must be initialized"
        self.group = group
        self.rank = dist.get_rank(group)
        self.size = dist.get_world_size(group)

    def barrier(self):
        dist.barrier(group=self.group)

    def bcast_(self, tensor: torch.Tensor, src: int, async_op: bool = False):
        return dist.broadcast(tensor, src=src, group=self.group,
                              async_op=async_op)

    def reduce_sum_(self, tensor: torch.Tensor, dst: int, async_op: bool = False):
        return dist.reduce(tensor, dst=dst, op=dist.ReduceOp.SUM,
                           group=self.group, async_op=async_op)

    def allreduce_sum_(self, tensor: torch.Tensor, async_op: bool = False):
        return dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group,
                               async_op=async_op)

    def allreduce_max_(self, tensor: torch.Tensor):
        # RCCL needs device tensors; round-trip CPU flags (e.g. the
        # fail-allreduce of arrow_bench) transparently
        if tensor.device.type == 'cpu' and dist.get_backend(self.group) == 'nccl':
            t = tensor.cuda()
            dist.all_reduce(t, op=dist.ReduceOp.MAX, group=self.group)
            tensor.copy_(t.cpu())
        else:
            dist.all_reduce(tensor, op=dist.ReduceOp.MAX, group=self.group)

    def alltoallv(self, send: torch.Tensor, send_counts: Sequence[int],
                  recv_counts: Sequence[int]) -> torch.Tensor:
        """Row-wise all-to-all-v: send rows [sum(send_counts[:r]) :
        +send_counts[r]] to rank r; returns recv tensor with
        sum(recv_counts) rows ordered by source rank.

        Implemented as grouped point-to-point (batch_isend_irecv): the RCCL
        alltoallv idiom over xGMI, and the only form gloo also supports.
        The self-block is copied locally (never hits the backend).
        CPU tensors under an nccl-only group (e.g. MatrixSlice's int
        tables) are round-tripped through the device.
        """
        assert send.dim() == 2
        if send.device.type == 'cpu' and dist.get_backend(self.group) == 'nccl':
            recv = self.alltoallv(send.cuda(), send_counts, recv_counts)
            return recv.cpu()
        if send.device.type == 'cuda' and dist.get_backend(self.group) == 'gloo':
            # gloo has no CUDA p2p: stage through host (validation rigs
            # only; the product multi-GPU path is RCCL)
            recv = self.alltoallv(send.cpu(), send_counts, recv_counts)
            return recv.to(send.device)
        sdispl = [0]
        for c in send_counts:
            sdispl.append(sdispl[-1] + int(c))
        rdispl = [0]
        for c in recv_counts:
            rdispl.append(rdispl[-1] + int(c))

        recv, works = self._post_alltoallv(send, send_counts, recv_counts,
                                           sdispl, rdispl)
        for w in works:
            w.wait()
        return recv

    def alltoallv_async(self, send: torch.Tensor, send_counts, recv_counts):
        """Post the exchange and return (recv, works) WITHOUT waiting —
        kernels enqueued before the waits overlap with the transfer
        (RCCL runs on its own stream; wait() inserts stream deps)."""
        if send.device.type == 'cuda' and dist.get_backend(self.group) == 'gloo':
            return self.alltoallv(send, send_counts, recv_counts), []
        sdispl = [0]
        for c in send_counts:
            sdispl.append(sdispl[-1] + int(c))
        rdispl = [0]
        for c in recv_counts:
            rdispl.append(rdispl[-1] + int(c))
        return self._post_alltoallv(send, send_counts, recv_counts, sdispl, rdispl)

    def _post_alltoallv(self, send, send_counts, recv_counts, sdispl, rdispl):
        k = send.shape[1]
        recv = send.new_empty((int(sum(recv_counts)), k))
        ops = []
        for r in range(self.size):
            if r == self.rank:
                continue
            if send_counts[r] > 0:
                ops.append(dist.P2POp(dist.isend,
                                      send[sdispl[r]:sdispl[r + 1]].contiguous(),
                                      r, group=self.group))
            if recv_counts[r] > 0:
                ops.append(dist.P2POp(dist.irecv, recv[rdispl[r]:rdispl[r + 1]],
                                      r, group=self.group))
        works = dist.batch_isend_irecv(ops) if ops else []
        # local block (current-stream device copy; overlaps with NCCL)
        if send_counts[self.rank] > 0:
            recv[rdispl[self.rank]:rdispl[self.rank + 1]].copy_(
                send[sdispl[self.rank]:sdispl[self.rank] + send_counts[self.rank]])
        return recv, works

    def allgather_cat(self, tensor: torch.Tensor) -> torch.Tensor:
        if tensor.device.type == 'cpu' and dist.get_backend(self.group) == 'nccl':
            return self.allgather_cat(tensor.cuda()).cpu()
        out: List[torch.Tensor] = [torch.empty_like(tensor) for _ in range(self.size)]
        dist.all_gather(out, tensor.contiguous(), group=self.group)
        return torch.cat(out, dim=0)

    def allgather_int(self, value: int) -> List[int]:
        t = torch.tensor([int(value)], dtype=torch.int64)
        return [int(x.item()) for x in self.allgather_cat(t)]

    def alltoall_ints(self, values: Sequence[int]) -> List[int]:
        """MPI Alltoall of one int per peer (matrix_slice.py:248)."""
        assert len(values) == self.size
        send = torch.tensor(values, dtype=torch.int64).view(-1, 1)
        recv = self.alltoallv(send, [1] * self.size, [1] * self.size)
        return [int(x.item()) for x in recv.view(-1)]


def default_comm() -> Comm:
    """TorchDistComm when torch.distributed is initialized, else the
    single-process comm."""
    if dist is not None and dist.is_available() and dist.is_initialized():
        return TorchDistComm()
    return Comm()

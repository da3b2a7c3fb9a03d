"""Ring communication layer — RCCL P2P over xGMI via torch.distributed.

MI355X-native re-design of the reference's ring layer
(``burst_attn/comm.py:104-321``), single-backend by design (north_star:
no multi-backend dispatch): ``torch.distributed`` with the ``nccl`` backend
IS RCCL on ROCm; on CPU test runs the same code drives ``gloo``.

Semantics kept from the reference:
  * per-round batched isend/irecv pairs to rank+1 / from rank-1 on the ring
    (``comm.py:148-172``), with the even/odd op ordering that avoids
    P2P deadlock (``comm.py:166-171``);
  * ``commit()`` → ``dist.batch_isend_irecv`` (``comm.py:269``);
    ``wait()`` blocks the compute stream on the transfer (``comm.py:301-321``).
    On ROCm, NCCL/RCCL P2P runs on the process group's internal HIP
    streams, so the transfer overlaps the attention kernel that is queued
    on the compute stream between commit() and wait() — the same overlap
    the reference gets from its side-stream commit (``comm.py:267-283``).

The two-level "double ring" (intra-node + inter-node groups,
``comm.py:187-254``) is multi-node-only — the reference itself disables it
single-node (``benchmarks/benchmark.py:41-44``) — and is not implemented in
this round (out of scope until multi-node exists).
Passing real double-ring groups raises NotImplementedError.
"""

import os

import torch
import torch.distributed as dist

__all__ = [
    "Ring",
    "replicate",
    "broadcast",
    "all_reduce",
    "synchronize",
    "gather_obj",
    "get_rank",
    "get_world_size",
    "get_local_world_size",
    "print_rank",
]


def replicate(tensor):
    """Out-of-place copy (reference ``comm.py:11-14``)."""
    res = torch.empty_like(tensor)
    res.copy_(tensor)
    return res


def broadcast(tensor, src, group=None):
    dist.broadcast(tensor, src, group)
    return tensor


def all_reduce(t, group=None):
    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return t


def get_world_size(group=None):
    return dist.get_world_size(group)


def get_rank(group=None):
    return dist.get_rank(group)


def get_local_world_size():
    return int(os.environ.get("LOCAL_WORLD_SIZE", 1))


def synchronize():
    if dist.is_initialized():
        dist.barrier()
    else:
        raise ValueError("Init torch.distributed first")


def gather_obj(obj):
    res = [None] * dist.get_world_size()
    dist.all_gather_object(res, obj)
    dist.barrier()
    return res


def print_rank(*args, **kwargs):
    if not dist.is_initialized() or dist.get_rank() == 0:
        print(*args, **kwargs)


class Ring:
    """Batched neighbour-ring P2P over one process group.

    Builds per-round op lists (``send_recv``), commits them in one
    ``batch_isend_irecv`` (one RCCL group call → one xGMI link each way),
    and ``wait()``s before the received buffers are consumed.

    API mirrors the reference ``Ring`` (``comm.py:104-321``) minus the
    bmtrain backend and (for now) the double ring.
    """

    def __init__(self, process_group=None, double_group=(None, None)):
        if double_group is not None and (
            double_group[0] is not None or double_group[1] is not None
        ):
            raise NotImplementedError(
                "double-ring (intra+inter node groups) is multi-node-only and "
                "not implemented yet; pass double_group=[None, None]"
            )
        self.comm = process_group
        self.world_size = dist.get_world_size(process_group)
        self.rank = dist.get_rank(process_group)
        self.local_group = None
        self.local_group2 = None
        self.intra_size = 1
        self.inter_size = 1
        self.ops = []
        self.reqs = []

    def _make_ring_ops(self, src_tensors, dst_tensors, group=None):
        comm = self.comm if group is None else group
        rank = dist.get_rank(comm)
        count = dist.get_world_size(comm)
        next_rank = (rank + 1) % count
        prev_rank = (rank - 1 + count) % count
        if comm is not None:
            next_rank = dist.get_global_rank(comm, next_rank)
            prev_rank = dist.get_global_rank(comm, prev_rank)
        ops = []
        for src, dst in zip(src_tensors, dst_tensors):
            send_op = dist.P2POp(dist.isend, src, next_rank, group=comm)
            recv_op = dist.P2POp(dist.irecv, dst, prev_rank, group=comm)
            # even/odd ordering avoids send/send head-of-line deadlock
            # (reference comm.py:166-171)
            if rank % 2 == 0:
                ops += [send_op, recv_op]
            else:
                ops += [recv_op, send_op]
        return ops

    def send_recv(self, tensor_list, dest_list):
        """Queue a ring hop: send each tensor to rank+1, receive the
        matching buffer from rank-1 (reference ``_ring_send_recv_base``,
        ``comm.py:256-257``).  A 1-rank ring is a local copy (gloo cannot
        send-to-self; semantically identical)."""
        if self.world_size == 1:
            for src, dst in zip(tensor_list, dest_list):
                dst.copy_(src)
            return
        self.ops += self._make_ring_ops(tensor_list, dest_list)

    # name kept so call sites read like the reference's
    # (double_ring_send_recv degenerates to the plain hop single-node,
    #  comm.py:221-227)
    def double_ring_send_recv(self, tensor_list, dest_list, r=0):
        self.send_recv(tensor_list, dest_list)

    def double_ring_send_recv_q(self, tensor_list, dest_list, r=0):
        self.send_recv(tensor_list, dest_list)

    def commit(self):
        if self.ops:
            self.reqs += dist.batch_isend_irecv(self.ops)
            self.ops = []

    def wait(self):
        for req in self.reqs:
            req.wait()
        self.reqs = []

"""Collective communication layer: RCCL over xGMI (via torch.distributed),
gloo for CPU tests.

The semantic contract implemented here is exactly the reference's shuffle
surface (SURVEY.md §5 "Distributed communication backend"): variable-size
all-to-all for hash/even/rand repartition, broadcast for small-side joins,
plus allgather of counts.  xGMI topology note: each MI355X has 7
point-to-point links (~153 GB/s each); the all-to-all-v is issued as
grouped P2P (``batch_isend_irecv`` → ``ncclGroupStart/Send/Recv``) so all
links run concurrently instead of a per-link-bound ring.
"""
import datetime
import os
from typing import Any, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

_COMM: Optional["Communicator"] = None


class Communicator:
    """Thin wrapper over a torch.distributed process group; world_size==1
    works without initialization."""

    def __init__(self, backend: Optional[str] = None, device: Optional[str] = None):
        if dist.is_available() and dist.is_initialized():
            self.world_size = dist.get_world_size()
            self.rank = dist.get_rank()
            self.backend = dist.get_backend()
        elif "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(
                backend=backend, timeout=datetime.timedelta(seconds=300)
            )
            self.world_size = dist.get_world_size()
            self.rank = dist.get_rank()
            self.backend = backend
        else:
            self.world_size = 1
            self.rank = 0
            self.backend = "none"
        self.device = device

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier()

    def allgather_counts(self, counts: torch.Tensor) -> torch.Tensor:
        """counts: [world_size] int64 of rows this rank sends to each peer.
        Returns matrix [world_size, world_size]: row r = what rank r sends."""
        if not self.is_distributed:
            return counts.reshape(1, -1).cpu()
        dev = self._comm_device(counts)
        local = counts.to(dev)
        gathered: List[torch.Tensor] = [
            torch.zeros_like(local) for _ in range(self.world_size)
        ]
        dist.all_gather(gathered, local)
        return torch.stack([t.cpu() for t in gathered], dim=0)

    def _comm_device(self, t: Optional[torch.Tensor] = None) -> torch.device:
        if self.backend == "nccl":
            if self.device is not None:
                return torch.device(self.device)
            # one process per GPU: LOCAL_RANK names this rank's device
            local = int(os.environ.get("LOCAL_RANK", "0"))
            return torch.device(
                f"cuda:{local % max(1, torch.cuda.device_count())}"
            )
        return torch.device("cpu")

    def all_to_all_v(
        self,
        send: torch.Tensor,
        send_counts: List[int],
        recv_counts: List[int],
    ) -> torch.Tensor:
        """Exchange variable-size 1-D slices of ``send`` (already
        bucket-contiguous, ordered by destination rank)."""
        if not self.is_distributed:
            return send
        # NCCL has no bool dtype: exchange as uint8 and cast back
        if send.dtype == torch.bool:
            out = self.all_to_all_v(
                send.to(torch.uint8), send_counts, recv_counts
            )
            return out.to(torch.bool)
        recv_total = sum(recv_counts)
        recv = torch.empty(
            recv_total, dtype=send.dtype, device=send.device
        )
        if self.backend == "nccl":
            dist.all_to_all_single(
                recv,
                send.contiguous(),
                output_split_sizes=recv_counts,
                input_split_sizes=send_counts,
            )
            return recv
        # gloo path: pairwise grouped isend/irecv (7-round all-pairs for 8
        # ranks; same schedule the NCCL backend lowers to on xGMI)
        send_offsets = [0]
        for c in send_counts:
            send_offsets.append(send_offsets[-1] + c)
        recv_offsets = [0]
        for c in recv_counts:
            recv_offsets.append(recv_offsets[-1] + c)
        # local copy
        recv[
            recv_offsets[self.rank] : recv_offsets[self.rank + 1]
        ] = send[send_offsets[self.rank] : send_offsets[self.rank + 1]]
        reqs = []
        for peer in range(self.world_size):
            if peer == self.rank:
                continue
            s = send[send_offsets[peer] : send_offsets[peer + 1]]
            r = recv[recv_offsets[peer] : recv_offsets[peer + 1]]
            if s.numel() > 0:
                reqs.append(dist.isend(s.contiguous(), dst=peer))
            if r.numel() > 0:
                reqs.append(dist.irecv(r, src=peer))
        for q in reqs:
            q.wait()
        return recv

    def broadcast_tensor(self, t: Optional[torch.Tensor], src: int, dtype=None, device=None) -> torch.Tensor:
        """Broadcast a 1-D tensor (shape+dtype negotiated via object bcast)."""
        if not self.is_distributed:
            assert t is not None
            return t
        meta: List[Any] = [None]
        if self.rank == src:
            assert t is not None
            meta = [(list(t.shape), str(t.dtype).replace("torch.", ""))]
        # NCCL transports object payloads through a device tensor: pin
        # the staging device explicitly so ranks with a non-default
        # current device can't deadlock on mismatched buffers
        if self.backend == "nccl":
            dist.broadcast_object_list(
                meta, src=src, device=self._comm_device(None)
            )
        else:
            dist.broadcast_object_list(meta, src=src)
        shape, dtype_name = meta[0]
        default_dev = (
            self._comm_device(None) if self.backend == "nccl" else "cpu"
        )
        if self.rank != src:
            t = torch.empty(
                shape,
                dtype=getattr(torch, dtype_name),
                device=device or default_dev,
            )
        dist.broadcast(t, src=src)
        return t

    def allreduce_sum(self, value: int) -> int:
        if not self.is_distributed:
            return value
        t = torch.tensor([value], dtype=torch.int64)
        t = t.to(self._comm_device(t))
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return int(t.cpu().item())


def get_communicator(device: Optional[str] = None) -> Communicator:
    global _COMM
    if _COMM is None:
        _COMM = Communicator(device=device)
    return _COMM


def reset_communicator() -> None:
    global _COMM
    _COMM = None

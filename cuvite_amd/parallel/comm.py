"""Distributed communication for one MI355X node: one process per GPU,
torch.distributed with RCCL ("nccl" on ROCm) over xGMI; gloo on CPU for tests.

Replaces the reference's MPI halo engine. Every GPU
pair on an MI355X node is directly connected by xGMI, so the var-size halo
exchanges map onto grouped point-to-point sends (ncclSend/ncclRecv via
torch.distributed.batch_isend_irecv); the tiny scalar reductions use a single
allreduce on a device buffer (no D2H round trip, unlike the reference's
per-iteration cudaMemcpy + MPI_Allreduce hop, louvain.cpp:2483-2537).
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist


class Comm:
    """Thin communicator. world_size 1 (no process group) short-circuits
    every operation, so single-process runs need no init."""

    def __init__(self, device: Optional[torch.device] = None,
                 transport: Optional[str] = None):
        self.active = dist.is_available() and dist.is_initialized()
        self.rank = dist.get_rank() if self.active else 0
        self.world = dist.get_world_size() if self.active else 1
        self.device = device if device is not None else torch.device("cpu")
        self.backend = dist.get_backend() if self.active else None
        # var-size exchange transport (ref compile-time matrix: Isend/Irecv
        # vs MPI_Alltoallv, louvain.cpp:2634-2682): "p2p" = grouped
        # batch_isend_irecv (default; all-pairs xGMI), "alltoall" = one
        # fused ncclAllToAllv-style call (nccl backend only; gloo has no
        # all_to_all). Runtime-selectable: CUVITE_TRANSPORT=alltoall.
        if transport is None:
            transport = os.environ.get("CUVITE_TRANSPORT", "p2p")
        if transport == "alltoall" and self.backend != "nccl":
            transport = "p2p"
        self.transport = transport

    # -------------------------------------------------------- collectives --

    def allreduce_sum_(self, t: torch.Tensor) -> torch.Tensor:
        if self.active and self.world > 1:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return t

    def allreduce_scalar(self, x: float, op: str = "sum") -> float:
        if not self.active or self.world == 1:
            return x
        t = torch.tensor([x], dtype=torch.float64, device=self.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max"
                        else dist.ReduceOp.SUM)
        return float(t[0])

    def allgather_counts(self, counts: torch.Tensor) -> torch.Tensor:
        """All-gather an int64 vector of per-peer counts; returns a
        [world, world] matrix M where M[p][q] = count p sends to q."""
        if not self.active or self.world == 1:
            return counts.reshape(1, -1)
        out = [torch.empty_like(counts) for _ in range(self.world)]
        dist.all_gather(out, counts.contiguous())
        return torch.stack(out)

    def barrier(self):
        if self.active and self.world > 1:
            dist.barrier()

    def gather_cat(self, t: torch.Tensor, root: int = 0) -> Optional[torch.Tensor]:
        """Gather variable-length 1-D tensors to root and concatenate in rank
        order (ref gatherAllComm, louvain.cpp:3306-3345). Cold path; routed
        through the same count-negotiated grouped-p2p machinery as the hot
        exchanges (all_to_all_v) so the first RCCL hardware contact has one
        code path, not a raw send/irecv variant on the side."""
        if not self.active or self.world == 1:
            return t
        empty = torch.empty(0, dtype=t.dtype, device=self.device)
        send = [t.contiguous() if p == root else empty
                for p in range(self.world)]
        got = self.all_to_all_v(send)
        if self.rank == root:
            return torch.cat([got[p] for p in range(self.world)])
        return None

    # ---------------------------------------------------------------- p2p --

    def all_to_all_v(self, send: List[torch.Tensor],
                     recv_counts: Optional[List[int]] = None) -> List[torch.Tensor]:
        """Variable-size all-to-all: send[p] goes to rank p; returns the list
        of tensors received from each rank. If recv_counts is None the counts
        are negotiated with an allgather first (ref halo size negotiation,
        louvain.cpp:2740). All tensors must share dtype; device buffers go
        straight over xGMI on the nccl/RCCL backend."""
        assert len(send) == self.world
        if not self.active or self.world == 1:
            return [send[0]]
        dtype = send[0].dtype
        me = self.rank
        if recv_counts is None:
            counts = torch.tensor([s.numel() for s in send], dtype=torch.int64,
                                  device=self.device)
            mat = self.allgather_counts(counts)
            recv_counts = [int(mat[p][me]) for p in range(self.world)]
        if self.transport == "alltoall":
            inp = torch.cat([send[p].contiguous() for p in range(self.world)])
            out = torch.empty(sum(recv_counts), dtype=dtype,
                              device=self.device)
            dist.all_to_all_single(
                out, inp, output_split_sizes=recv_counts,
                input_split_sizes=[int(s.numel()) for s in send])
            return list(torch.split(out, recv_counts))
        recv = [torch.empty(recv_counts[p], dtype=dtype, device=self.device)
                for p in range(self.world)]
        ops = []
        for p in range(self.world):
            if p == me:
                recv[p] = send[p]
                continue
            if send[p].numel() > 0:
                ops.append(dist.P2POp(dist.isend, send[p].contiguous(), p))
            if recv_counts[p] > 0:
                ops.append(dist.P2POp(dist.irecv, recv[p], p))
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        return recv

    def exchange_fixed(self, send: List[torch.Tensor],
                       recv: List[torch.Tensor]):
        """Pre-negotiated exchange into preallocated receive buffers (the per-
        iteration ghost-label exchange: sizes are static within a phase)."""
        if not self.active or self.world == 1:
            if recv and recv[0].numel():
                recv[0].copy_(send[0])
            return
        me = self.rank
        ops = []
        for p in range(self.world):
            if p == me:
                if recv[p].numel():
                    recv[p].copy_(send[p])
                continue
            if send[p].numel() > 0:
                ops.append(dist.P2POp(dist.isend, send[p].contiguous(), p))
            if recv[p].numel() > 0:
                ops.append(dist.P2POp(dist.irecv, recv[p], p))
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()


def init_from_env(backend: Optional[str] = None,
                  timeout_s: int = 600, prefer: str = "auto") -> Comm:
    """Initialize torch.distributed from torchrun env vars and pick the GPU
    for this rank (ref set_gpuDevices, louvain_cuda.cu:1634-1669: one device
    per node-local rank). prefer: "auto" (GPU if present), "cpu", "cuda"."""
    use_cuda = torch.cuda.is_available() if prefer == "auto" \
        else prefer == "cuda"
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world == 1:
        dev = torch.device("cuda:0") if use_cuda else torch.device("cpu")
        if dev.type == "cuda":
            torch.cuda.set_device(dev)
        return Comm(dev)
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = torch.device("cuda", local_rank % torch.cuda.device_count())
    else:
        device = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_s))
    return Comm(device)

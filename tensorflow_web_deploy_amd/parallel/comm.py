"""Collective-comm backend for DP serving:

  weights:  broadcast rank0 -> all (once, PCIe H2D then xGMI fan-out)
  requests: scatter super-batch shards rank0 -> all   ("request scatter")
  results:  gather logits all -> rank0                ("result gather")

`TorchDistComm` wraps torch.distributed — backend "nccl" IS RCCL on
ROCm, running over the node's fully-connected xGMI mesh (7 p2p links/GPU;
scatter/gather are single-hop sends, no ring needed). The same code runs
under "gloo" on CPU for tests. `LoopbackComm` is the in-process fake
 used to unit-test sharding math with no process group.
"""
from __future__ import annotations

from typing import List, Optional

import torch


def shard_sizes(total: int, world: int) -> List[int]:
    """Split `total` rows into `world` contiguous shards (front-loaded)."""
    base, rem = divmod(total, world)
    return [base + (1 if r < rem else 0) for r in range(world)]


class TorchDistComm:
    """torch.distributed process-group comm (RCCL on GPU, gloo on CPU)."""

    def __init__(self, rank: int, world: int, device: torch.device):
        import torch.distributed as dist
        self.dist = dist
        self.rank = rank
        self.world = world
        self.device = device

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        self.dist.broadcast(t, src=src)
        return t

    def scatter_batch(self, batch: Optional[torch.Tensor],
                      shape_tail: tuple, dtype=torch.float32) -> torch.Tensor:
        """rank0 passes the full batch; others pass None. Returns the local
        shard. Batch rows are padded so every rank gets an equal shard
        (required by collective scatter); callers slice via sizes."""
        n = torch.tensor([batch.shape[0] if batch is not None else 0],
                         dtype=torch.int64)
        if self.device.type != "cpu":
            n = n.to(self.device)
        self.dist.broadcast(n, src=0)
        total = int(n.item())
        per = (total + self.world - 1) // self.world
        local = torch.empty((per, *shape_tail), dtype=dtype,
                            device=self.device)
        if self.rank == 0:
            padded = torch.zeros((per * self.world, *shape_tail), dtype=dtype,
                                 device=self.device)
            padded[:total] = batch.to(self.device)
            chunks = list(padded.chunk(self.world, dim=0))
            self.dist.scatter(local, chunks, src=0)
        else:
            self.dist.scatter(local, None, src=0)
        # how many of my rows are real
        start = self.rank * per
        mine = max(0, min(per, total - start))
        return local[:mine]

    def gather_rows(self, local: torch.Tensor, total: int,
                    shape_tail: tuple) -> Optional[torch.Tensor]:
        """Inverse of scatter_batch: gather shards to rank0, return the
        concatenated [total, ...] tensor on rank0 (None elsewhere)."""
        per = (total + self.world - 1) // self.world
        send = torch.zeros((per, *shape_tail), dtype=local.dtype,
                           device=self.device)
        send[:local.shape[0]] = local
        if self.rank == 0:
            bufs = [torch.empty_like(send) for _ in range(self.world)]
            self.dist.gather(send, bufs, dst=0)
            out = torch.cat(bufs, dim=0)[:total]
            return out
        self.dist.gather(send, None, dst=0)
        return None

    def barrier(self):
        self.dist.barrier()


def broadcast_graph(graph, rank: int, device: torch.device, dist):
    """Rank0 fans the loaded model out to every replica (SURVEY §2b row
    "model load"): ONE disk read on rank0, then (1) the graph skeleton +
    const metadata via broadcast_object_list, (2) all const/weight bytes
    packed into a single flat tensor broadcast once — on the nccl(=RCCL)
    backend that is one xGMI fan-out (~51 MB for bf16 ResNet-50) instead
    of 8 independent disk loads / random inits.

    Rank0 passes its Graph and gets it back unchanged; other ranks pass
    None and receive a fully-populated copy.
    """
    import numpy as np

    from ..engine.ir import Graph
    if rank == 0:
        names = sorted(graph.consts)
        metas = [(n, graph.consts[n].dtype.str,
                  tuple(graph.consts[n].shape)) for n in names]
        skeleton = Graph(graph.nodes, {}, graph.inputs, graph.outputs)
        obj = [skeleton, metas]
    else:
        obj = [None, None]
    dist.broadcast_object_list(obj, src=0)
    skeleton, metas = obj
    sizes = [int(np.prod(s)) * np.dtype(d).itemsize if len(s)
             else np.dtype(d).itemsize for (_, d, s) in metas]
    total = int(sum(sizes))
    if total == 0:
        return graph if rank == 0 else skeleton
    buf = torch.empty(total, dtype=torch.uint8, device=device)
    if rank == 0:
        packed = np.empty(total, dtype=np.uint8)
        off = 0
        for (n, _, _), nb in zip(metas, sizes):
            packed[off:off + nb] = np.frombuffer(
                np.ascontiguousarray(graph.consts[n]).tobytes(), np.uint8)
            off += nb
        buf.copy_(torch.from_numpy(packed))
    dist.broadcast(buf, src=0)
    if rank == 0:
        return graph
    host = buf.cpu().numpy()
    consts = {}
    off = 0
    for (n, d, s), nb in zip(metas, sizes):
        consts[n] = host[off:off + nb].copy().view(np.dtype(d)).reshape(s)
        off += nb
    skeleton.consts = consts
    return skeleton


class LoopbackComm:
    """In-process world_size=1 fake with the same interface (CPU tests of
    the replica manager / sharding math without a process group)."""

    def __init__(self):
        self.rank = 0
        self.world = 1
        self.device = torch.device("cpu")

    def broadcast(self, t, src=0):
        return t

    def scatter_batch(self, batch, shape_tail, dtype=torch.float32):
        return batch

    def gather_rows(self, local, total, shape_tail):
        return local[:total]

    def barrier(self):
        pass

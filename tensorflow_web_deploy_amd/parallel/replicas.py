"""DP replica manager: one process per GPU, RCCL scatter/gather serving.

Architecture:
  server process (Flask + batcher)
    └── rank-0 worker: receives super-batches over a shm queue, uploads,
        RCCL-scatters shards over xGMI, runs its own shard, gathers
        probs, returns them to the server.
    └── ranks 1..N-1: scatter -> forward -> gather loop.

Health: each worker heartbeats via its process liveness; a dead replica
triggers group shutdown and a respawn with the survivors (shrink-serve),
see `ReplicatedEngine._respawn`.

The same code runs with backend=gloo + device=cpu for the world>1 CPU
tests (no GPU box has >1 GPU available to us; the 8-GPU path is
exercised by the driver's scaling bench).
"""
from __future__ import annotations

import os
import queue
import time
import traceback
from typing import Optional

import numpy as np
import torch
import torch.multiprocessing as mp

from ..conf.config import Config
from .comm import TorchDistComm


def _pick_port() -> int:
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def worker_main(rank: int, world: int, cfg: Config, backend: str, port: int,
                in_q, out_q, err_q):
    """Replica worker entry (separate process)."""
    try:
        import torch.distributed as dist
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        if backend == "nccl":
            torch.cuda.set_device(rank)
            device = torch.device(f"cuda:{rank}")
        else:
            device = torch.device("cpu")
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
        comm = TorchDistComm(rank, world, device)

        from ..engine.service import LocalEngine
        eng = LocalEngine(cfg.replace(device=str(device)))

        h, w, c = cfg.input_height, cfg.input_width, cfg.input_channels
        nc = cfg.num_classes
        if rank == 0:
            out_q.put(("ready", world))

        while True:
            if rank == 0:
                item = in_q.get()
                if item is None:
                    n0 = torch.tensor([-1], dtype=torch.int64, device=device
                                      if backend == "nccl" else "cpu")
                    dist.broadcast(n0, src=0)
                    break
                batch = torch.from_numpy(item)
            else:
                batch = None
            # scatter_batch broadcasts the row count first; -1 = shutdown
            if rank != 0:
                n0 = torch.tensor([0], dtype=torch.int64, device=device
                                  if backend == "nccl" else "cpu")
                dist.broadcast(n0, src=0)
                total = int(n0.item())
                if total < 0:
                    break
                shard = comm._scatter_known(total, (h, w, c))
            else:
                total = batch.shape[0]
                n0 = torch.tensor([total], dtype=torch.int64, device=device
                                  if backend == "nccl" else "cpu")
                dist.broadcast(n0, src=0)
                shard = comm._scatter_known(total, (h, w, c), batch)
            if shard.shape[0] > 0:
                probs = eng.ex.run({cfg.input_name: shard})[0].float()
            else:
                probs = torch.zeros((0, nc), dtype=torch.float32,
                                    device=device)
            out = comm.gather_rows(probs, total, (nc,))
            if rank == 0:
                out_q.put(("probs", out.cpu().numpy()))
        dist.destroy_process_group()
    except Exception:
        err_q.put((rank, traceback.format_exc()))
        raise


def _scatter_known(self, total, shape_tail, batch=None):
    """Scatter when the row count is already broadcast (method injected
    into TorchDistComm to keep the wire protocol in one place)."""
    per = (total + self.world - 1) // self.world
    local = torch.empty((per, *shape_tail), dtype=torch.float32,
                        device=self.device)
    if self.rank == 0:
        padded = torch.zeros((per * self.world, *shape_tail),
                             dtype=torch.float32, device=self.device)
        padded[:total] = batch.to(self.device)
        self.dist.scatter(local, list(padded.chunk(self.world, 0)), src=0)
    else:
        self.dist.scatter(local, None, src=0)
    start = self.rank * per
    mine = max(0, min(per, total - start))
    return local[:mine]


TorchDistComm._scatter_known = _scatter_known


class ReplicatedEngine:
    """`infer(batch) -> probs` over N DP replica processes.

    Failure policy (SURVEY §5 "failure detection"): a dead replica is
    detected on the next call; with shrink_on_failure the group is torn
    down and respawned with world-1 (8 -> 7 -> ...) and the in-flight
    batch is retried on the survivors."""

    def __init__(self, cfg: Config, world: Optional[int] = None,
                 backend: Optional[str] = None, start_timeout: float = 600.0,
                 shrink_on_failure: bool = True):
        self.cfg = cfg
        self.world = world or cfg.gpus
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.backend = backend
        self.shrink_on_failure = shrink_on_failure
        self._start_timeout = start_timeout
        self._spawn(start_timeout)

    def _spawn(self, timeout: float):
        ctx = mp.get_context("spawn")
        self.in_q = ctx.Queue()
        self.out_q = ctx.Queue()
        self.err_q = ctx.Queue()
        port = _pick_port()
        self.procs = []
        for r in range(self.world):
            p = ctx.Process(target=worker_main,
                            args=(r, self.world, self.cfg, self.backend,
                                  port, self.in_q, self.out_q, self.err_q),
                            daemon=True)
            p.start()
            self.procs.append(p)
        # wait for readiness
        deadline = time.time() + timeout
        while True:
            try:
                tag, _ = self.out_q.get(timeout=1.0)
                assert tag == "ready"
                break
            except queue.Empty:
                self._check_health()
                if time.time() > deadline:
                    self.shutdown()
                    raise TimeoutError("replicas failed to start")

    def _check_health(self):
        try:
            rank, tb = self.err_q.get_nowait()
            self._teardown()
            raise RuntimeError(f"replica {rank} died:\n{tb}")
        except queue.Empty:
            pass
        for r, p in enumerate(self.procs):
            if not p.is_alive():
                self._teardown()
                raise RuntimeError(f"replica {r} exited unexpectedly")

    def _teardown(self):
        for p in self.procs:
            if p.is_alive():
                p.terminate()
        for p in self.procs:
            p.join(timeout=5.0)

    def infer(self, batch: np.ndarray) -> np.ndarray:
        batch = np.ascontiguousarray(batch, dtype=np.float32)
        try:
            return self._infer_once(batch)
        except RuntimeError:
            if not self.shrink_on_failure or self.world <= 1:
                raise
            # shrink-serve: respawn with the survivors and retry
            self.shutdown()
            self.world -= 1
            self._spawn(self._start_timeout)
            return self._infer_once(batch)

    def _infer_once(self, batch: np.ndarray) -> np.ndarray:
        self.in_q.put(batch)
        while True:
            try:
                tag, payload = self.out_q.get(timeout=5.0)
                assert tag == "probs"
                return payload
            except queue.Empty:
                self._check_health()

    def shutdown(self):
        try:
            self.in_q.put(None)
        except Exception:
            pass
        for p in self.procs:
            p.join(timeout=10.0)
        for p in self.procs:
            if p.is_alive():
                p.terminate()

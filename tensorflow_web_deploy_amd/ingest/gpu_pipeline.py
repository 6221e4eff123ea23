"""GPU ingest pipeline: decode on CPU, resize/normalize on a side HIP
stream (north star: "JPEG decode and NHWC preprocess run on a side HIP
stream into 288 GB HBM per GPU").

Division of labor:
  * JPEG/PNG entropy decode is inherently serial -> CPU (PIL's C path,
    GIL-released), producing uint8 RGB at the original size.
  * Everything parallel — bilinear resize, normalization, NHWC pack —
    runs as a HIP kernel (csrc/preprocess.hip) on a dedicated side
    stream, overlapped with the main stream's model compute; an event
    hands the batch to the compute stream.

Validated against the PIL correctness path (ingest/preprocess.py) in
tests/test_ingest_gpu.py.
"""
from __future__ import annotations

import io
from concurrent.futures import ProcessPoolExecutor
from typing import List, Optional, Sequence

import numpy as np
import torch


def _pool_init():
    import gc
    gc.collect()
    gc.freeze()
    gc.set_threshold(200000, 100, 100)


def _decode_worker(args):
    """Module-level for process-pool pickling: bytes -> uint8 HWC."""
    data, channels = args
    from PIL import Image
    img = Image.open(io.BytesIO(data))
    img = img.convert("L" if channels == 1 else "RGB")
    arr = np.asarray(img, dtype=np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    return arr


# --- shared-memory decode pool ---------------------------------------
# Measured (round 1): pool.map that returns decoded arrays costs ~56 ms
# of pure IPC per 64-image batch (72 ms round-trip vs 52 ms SERIAL
# decode), and ProcessPoolExecutor dispatch ALONE costs 20-40 ms per 64
# no-op tasks. So this pool uses neither: persistent worker processes
# share one mmap'd /dev/shm file (spawn children reattach by path — no
# resource_tracker involvement). Per batch the parent writes a task
# table into the mapping and wakes each worker with ONE pipe byte;
# worker j decodes tasks j, j+K, ... straight into fixed-size pixel
# slots and answers with one byte. Only 2 syscalls per worker per
# batch; blobs out and pixels back never touch a pipe. The parent's
# view of the slot area is hipHostRegister'ed once so the H2D of the
# slot prefix is a true pinned-memory DMA.
#
# Mapping layout:
#   [task table: slots x 2 int64 (in_off, in_len)]
#   [meta table: slots x 4 int32 (h, w, c, status)]
#   [input area: packed request blobs]
#   [slot area:  slots x slot_bytes decoded pixels]

_SHM = {"mm": None, "path": None}


def _shm_attach(path):
    import mmap
    f = open(path, "r+b")
    mm = mmap.mmap(f.fileno(), 0)
    _SHM["mm"] = np.frombuffer(mm, dtype=np.uint8)
    _SHM["_keep"] = (f, mm)
    return _SHM["mm"]


JPEG_MAGIC = 0x4A504731  # SlotHdr.magic (csrc/jpeg_gpu.hip)
JPEG_HDR_BYTES = 512     # pixels land at slot+512 (header stays intact)


def _write_jpeg_slot(buf, off, slot_bytes, info):
    """Serialize a _jpegc.parse() result as a GPU coeff slot (SlotHdr
    layout in csrc/jpeg_gpu.hip). Returns (H, W, 3) or None if it does
    not fit (caller falls back to pixel decode)."""
    W, H, nc = info["width"], info["height"], info["ncomp"]
    if nc != 3:
        return None
    hdr = np.zeros(23, np.int32)
    hdr[0] = JPEG_MAGIC
    hdr[1], hdr[2], hdr[3] = W, H, nc
    coef_off, plane_off = JPEG_HDR_BYTES, 0
    for ci in range(nc):
        bh, bw = info["blocks"][ci]
        h_, v_ = info["sampling"][ci]
        hdr[4 + ci], hdr[7 + ci] = h_, v_
        hdr[10 + ci], hdr[13 + ci] = bh, bw
        hdr[16 + ci], hdr[19 + ci] = coef_off, plane_off
        coef_off += bh * bw * 128
        plane_off += bh * 8 * bw * 8
    if (coef_off > slot_bytes or plane_off > slot_bytes or
            JPEG_HDR_BYTES + W * H * 3 > slot_bytes):
        return None
    buf[off:off + 92] = hdr.view(np.uint8)
    qts = np.concatenate([np.asarray(info["qt"][ci], np.uint16)
                          for ci in range(nc)])
    buf[off + 92:off + 92 + 384] = qts.view(np.uint8)
    for ci in range(nc):
        cb = np.asarray(info["coef"][ci], np.int16).view(np.uint8)
        o = off + int(hdr[16 + ci])
        buf[o:o + cb.nbytes] = cb
    return (H, W, 3)


def _ring_worker(path, widx, nworkers, channels, slot_bytes, slots,
                 in_bytes, wake_rx, done_tx, jpeg=False):
    """Persistent decode worker: wake byte -> decode my stripe -> done
    byte. Runs in its own process; PIL decode releases the GIL there.
    With jpeg=True, baseline JPEGs are entropy-decoded only (_jpegc)
    and written as COEFF slots (status 3) for the GPU IDCT kernels;
    anything else falls back to full PIL pixel decode."""
    import sys
    import traceback

    def _die(exc):
        traceback.print_exc(file=sys.stderr)
        sys.stderr.flush()
        raise exc

    import gc
    gc.collect()
    gc.freeze()
    from PIL import Image
    jparse = None
    if jpeg and channels == 3:
        try:
            from tensorflow_web_deploy_amd import _jpegc
            jparse = _jpegc.parse
        except Exception:
            jparse = None
    buf = _shm_attach(path)
    tasks = np.frombuffer(buf[:slots * 16], dtype=np.int64)
    meta = np.frombuffer(buf[slots * 16:slots * 16 + slots * 16],
                         dtype=np.int32)
    base_in = slots * 32
    base_out = base_in + in_bytes
    while True:
        try:
            msg = wake_rx.recv_bytes()
        except EOFError:
            return
        if msg == b"q":
            return
        n = int(tasks[0])   # entry 0: task count
        sbase = int(tasks[1])  # entry 1: slot-index base (phase half)
        for i in range(widx, n, nworkers):
            in_off = int(tasks[2 * (i + 1)])
            in_len = int(tasks[2 * (i + 1) + 1])
            data = bytes(buf[base_in + in_off:base_in + in_off + in_len])
            if jparse is not None:
                try:
                    info = jparse(data)
                    sh = _write_jpeg_slot(
                        buf, base_out + (sbase + i) * slot_bytes,
                        slot_bytes, info)
                    if sh is not None:
                        meta[4 * i:4 * i + 4] = (sh[0], sh[1], sh[2], 3)
                        continue
                except (ValueError, RuntimeError):
                    # invalid_argument (progressive/CMYK/...) or runtime_error
                    # (corrupt entropy stream): PIL pixel path below
                    pass
            try:
                img = Image.open(io.BytesIO(data))
                img = img.convert("L" if channels == 1 else "RGB")
                arr = np.asarray(img, dtype=np.uint8)
                if arr.ndim == 2:
                    arr = arr[:, :, None]
                if arr.nbytes > slot_bytes:
                    meta[4 * i:4 * i + 4] = (0, 0, 0, 2)  # too big
                    continue
                o = base_out + (sbase + i) * slot_bytes
                buf[o:o + arr.nbytes] = arr.reshape(-1)
                meta[4 * i:4 * i + 4] = (arr.shape[0], arr.shape[1],
                                         arr.shape[2], 0)
            except Exception:
                meta[4 * i:4 * i + 4] = (0, 0, 0, 1)  # bad image
        try:
            done_tx.send_bytes(b"d")
        except (BrokenPipeError, OSError):
            return




class ShmDecodePool:
    """Decode JPEG/PNG blobs into fixed-size shared-memory slots.

    decode_batch() returns int32 meta rows (slot_byte_offset, h, w, c)
    relative to the slot area, directly consumable by
    resize_norm_batch; returns None when the batch doesn't fit the
    layout or a worker died (caller uses the legacy pipe path);
    raises ValueError naming the first undecodable image.
    """

    def __init__(self, channels: int, workers: Optional[int] = None,
                 slot_bytes: int = 1 << 19, slots: int = 256,
                 in_bytes: int = 64 << 20, jpeg: bool = False,
                 phases: int = 2):
        import multiprocessing as mp
        import os
        import tempfile
        self.c = channels
        self.jpeg = jpeg
        self.slot_bytes = slot_bytes
        self.slots = slots
        self.in_bytes = in_bytes
        self.hdr = slots * 32
        total = self.hdr + in_bytes + slots * slot_bytes
        # slot area is N-way buffered: a pipelined server decodes batch
        # N+k into a free window while batch N's window is still being
        # DMA'd / consumed by the GPU stage (window tokens gate reuse;
        # >2 phases keep the decode stage from stalling on the forward)
        self.phases = max(2, phases)
        self.half = (slots - 2) // self.phases  # slots per window
        self._phase = 0
        self.last_window_off = 0  # byte offset of the active half
        fd, self.path = tempfile.mkstemp(prefix="twd_dec_",
                                         dir="/dev/shm")
        os.ftruncate(fd, total)
        os.close(fd)
        self.buf = _shm_attach(self.path)
        self.buf[:: 4096] = 0  # pre-touch: fault every page once, now
        self._tasks = np.frombuffer(self.buf[:slots * 16], dtype=np.int64)
        self._meta = np.frombuffer(
            self.buf[slots * 16:slots * 32], dtype=np.int32)
        if workers is None:
            # 48-worker cap A/B'd +3% over 32 at 256 clients (the
            # load generator shares the box; more workers win the
            # core-stealing fight) — profiles/round2_serving_ab.md
            workers = min(48, max(4, (os.cpu_count() or 8) - 2))
        self.workers = workers
        ctx = mp.get_context("spawn")
        self._wake, self._done, self._procs = [], [], []
        for j in range(workers):
            wr, ww = ctx.Pipe(duplex=False)   # parent -> worker
            dr, dw = ctx.Pipe(duplex=False)   # worker -> parent
            pr = ctx.Process(
                target=_ring_worker,
                args=(self.path, j, workers, channels, slot_bytes,
                      slots, in_bytes, wr, dw, jpeg),
                daemon=True)
            pr.start()
            wr.close()
            dw.close()
            self._wake.append(ww)
            self._done.append(dr)
            self._procs.append(pr)
        self._dead = False
        self._warm = False
        # /dev/shm files outlive the process: without this a crashed or
        # un-close()d server leaks a ~200-300 MB mapping per run
        import atexit
        atexit.register(self.close)

    def warm(self, timeout: float = 60.0):
        if self._warm:
            return
        from PIL import Image
        b = io.BytesIO()
        Image.new("RGB" if self.c == 3 else "L", (8, 8)).save(b, "PNG")
        self.decode_batch([b.getvalue()] * min(self.slots,
                                               2 * self.workers),
                          timeout=timeout)
        self._warm = True

    def decode_batch(self, blobs, timeout: float = 30.0,
                     phase: Optional[int] = None):
        """phase: which slot half to decode into (a pipelined caller
        owns the handoff; None = internal alternation). NOT
        thread-safe — one decoding thread at a time."""
        if self._dead:
            return None
        n = len(blobs)
        if (n == 0 or n > self.half or
                sum(len(b) for b in blobs) > self.in_bytes):
            return None
        if phase is None:
            phase = self._phase
            self._phase = (self._phase + 1) % self.phases
        self.last_window_off = phase * self.half * self.slot_bytes
        self._tasks[0] = n
        self._tasks[1] = phase * self.half  # slot index base for workers
        off = 0
        base_in = self.hdr
        for i, b in enumerate(blobs):
            lb = len(b)
            self.buf[base_in + off:base_in + off + lb] = \
                np.frombuffer(b, np.uint8)
            self._tasks[2 * (i + 1)] = off
            self._tasks[2 * (i + 1) + 1] = lb
            off += lb
        nw = min(self.workers, n)
        for j in range(nw):
            self._wake[j].send_bytes(b"g")
        try:
            for j in range(nw):
                if not self._done[j].poll(timeout):
                    import sys
                    self._dead = True  # worker gone: don't trust the ring
                    print(f"[ShmDecodePool] worker {j} timeout "
                          f"({timeout}s); falling back", file=sys.stderr,
                          flush=True)
                    return None
                self._done[j].recv_bytes()
        except (EOFError, OSError) as e:
            import sys
            self._dead = True
            dead = [j for j, pr in enumerate(self._procs)
                    if not pr.is_alive()]
            codes = [self._procs[j].exitcode for j in dead]
            print(f"[ShmDecodePool] worker pipe error {e!r}; dead "
                  f"workers {dead} exitcodes {codes}; falling back to "
                  f"the pipe decode path", file=sys.stderr, flush=True)
            return None
        meta = np.empty((n, 4), np.int32)
        flags = np.zeros(n, np.uint8)
        m = self._meta
        for i in range(n):
            st = m[4 * i + 3]
            if st == 1:
                raise ValueError(f"cannot decode image {i}")
            if st == 2:
                return None  # exceeds a slot: legacy path handles it
            # offsets RELATIVE to the slot area: the H2D copies the
            # slot prefix only, so the kernel sees that base. Status 3
            # = COEFF slot (GPU jpeg path): pixels will appear at +512
            # after the jpeg kernels run.
            off = i * self.slot_bytes
            if st == 3:
                off += JPEG_HDR_BYTES
                flags[i] = 1
            meta[i] = (off, m[4 * i], m[4 * i + 1], m[4 * i + 2])
        self.last_flags = flags
        return meta

    def close(self):
        import os
        if getattr(self, "_closed", False):
            return
        self._closed = True
        for w in self._wake:
            try:
                w.send_bytes(b"q")
            except (BrokenPipeError, OSError):
                pass
        for pr in self._procs:
            pr.join(timeout=1.0)
            if pr.is_alive():
                pr.terminate()
        try:
            os.unlink(self.path)
        except OSError:
            pass


class GpuPreprocessor:
    def __init__(self, height: int, width: int, channels: int,
                 normalize: str = "unit", device: str = "cuda",
                 max_batch: int = 256):
        from ..ops.native import ext
        self.E = ext()
        self.h, self.w, self.c = height, width, channels
        self.norm = {"raw": 0, "unit": 1, "inception": 2}[normalize]
        self.device = torch.device(device)
        self.stream = torch.cuda.Stream(device=self.device)  # side stream
        self.max_batch = max_batch
        self._out = torch.empty((max_batch, height, width, channels),
                                dtype=torch.float32, device=self.device)
        self._pin: dict = {}
        # Decode in WORKER PROCESSES: PIL in server threads starves the
        # batcher thread for the GIL under load (measured: 90-400 ms
        # batch stalls); processes sidestep the GIL entirely.
        import multiprocessing as mp
        self._pool = ProcessPoolExecutor(
            max_workers=8, mp_context=mp.get_context("spawn"),
            initializer=_pool_init)
        self._pool_warm = False
        self._shm_pool: Optional[ShmDecodePool] = None
        self._shm_t = None
        self._shm_pinned = False
        self._planes = None  # GPU scratch for the jpeg IDCT kernels
        import os as _os
        # pipeline depth: decode windows + per-phase GPU output batches
        # (3 = decode can run two batches ahead of the forward)
        self.phases = max(2, int(_os.environ.get("TWD_PIPE_PHASES", "3")))
        self._out_phases = [self._out]  # lazily extended per phase

    def _decode_cpu(self, data: bytes) -> np.ndarray:
        from PIL import Image
        img = Image.open(io.BytesIO(data))
        img = img.convert("L" if self.c == 1 else "RGB")
        arr = np.asarray(img, dtype=np.uint8)
        if arr.ndim == 2:
            arr = arr[:, :, None]
        return arr

    def process(self, blobs: Sequence[bytes],
                compute_stream: Optional[torch.cuda.Stream] = None
                ) -> torch.Tensor:
        """blobs -> f32 NHWC batch tensor on the GPU (model-input scale).

        Upload + resize/normalize run on the side stream; the returned
        tensor is safe to consume on `compute_stream` (event-synced).
        """
        if len(blobs) > 4:
            out = self.process_shm(blobs, compute_stream)
            if out is not None:
                return out
        if len(blobs) <= 4:
            # small batches: pool round-trip costs more than decode
            arrs = [_decode_worker((b, self.c)) for b in blobs]
        else:
            arrs = list(self._pool.map(_decode_worker,
                                       [(b, self.c) for b in blobs],
                                       chunksize=4))
        return self.process_arrays(arrs, compute_stream)

    def _ensure_shm(self):
        if self._shm_pool is None:
            import os
            nw = os.environ.get("TWD_DECODE_WORKERS")
            # half-size (slots-2)//2 must cover max_batch: each
            # pipeline phase decodes a full batch into one half
            import os as _os
            jpeg = (_os.environ.get("TWD_GPU_JPEG", "1") != "0"
                    and self.c == 3)
            # slot size bounds the largest decodable image (h*w*c bytes)
            # AND the per-batch H2D volume (n * slot_bytes): 512 KiB
            # fits ~400x400x3 but ships 3.4x padding for 224^2 uploads
            sb = int(os.environ.get("TWD_SLOT_BYTES", 1 << 19))
            self._shm_pool = ShmDecodePool(
                self.c, workers=int(nw) if nw else None,
                slot_bytes=sb,
                slots=max(self.phases * 129 + 2,
                          self.phases * self.max_batch + 2),
                jpeg=jpeg, phases=self.phases)
            self._shm_t = torch.from_numpy(self._shm_pool.buf)
            self._shm_pinned = bool(self.E.host_register(
                self._shm_t.data_ptr(), self._shm_t.numel()))
        return self._shm_pool

    def decode_shm(self, blobs, phase=None):
        """CPU stage: ring-decode into the active slot half. Returns
        (meta, window_byte_off) or None (caller uses the pipe path);
        raises ValueError for an undecodable image."""
        pool = self._ensure_shm()
        meta = pool.decode_batch(blobs, phase=phase)
        if meta is None:
            return None
        return meta, pool.last_window_off, pool.last_flags

    def _out_buf(self, phase):
        """Per-phase output batches: the pipelined server uploads batch
        N+k (side stream) while batch N's forward still reads its own
        output tensor; phase tokens gate reuse."""
        if phase is None:
            phase = 0
        while len(self._out_phases) <= phase:
            self._out_phases.append(torch.empty_like(self._out))
        return self._out_phases[phase]

    def upload_shm(self, decoded, n: int,
                   compute_stream: Optional[torch.cuda.Stream] = None,
                   phase: Optional[int] = None, sync: bool = True):
        """GPU stage: one pinned DMA of the slot window (+ the jpeg
        dequant/IDCT/color kernels for coeff slots) + one resize kernel,
        all on the side stream. sync=False returns (out, event) with NO
        wait enqueued — the consumer thread waits on the event itself
        (cross-thread default-stream waits would order arbitrarily)."""
        meta, win, flags = decoded
        pool = self._shm_pool
        out = self._out_buf(phase)[:n]
        lo = pool.hdr + pool.in_bytes + win
        src = self._shm_t[lo:lo + n * pool.slot_bytes]
        with torch.cuda.stream(self.stream):
            arena_gpu = src.to(self.device, non_blocking=True)
            if flags is not None and flags.any():
                if self._planes is None:
                    self._planes = torch.empty(
                        (pool.half + 1) * pool.slot_bytes,
                        dtype=torch.uint8, device=self.device)
                fl = torch.from_numpy(flags).to(self.device,
                                                non_blocking=True)
                self.E.jpeg_decode_slots(arena_gpu, self._planes, fl,
                                         n, pool.slot_bytes)
            meta_gpu = torch.from_numpy(meta).to(self.device,
                                                 non_blocking=True)
            self.E.resize_norm_batch(arena_gpu, meta_gpu, out, self.norm)
        ev = torch.cuda.Event()
        ev.record(self.stream)
        if not sync:
            return out, ev
        target = compute_stream or torch.cuda.current_stream()
        target.wait_event(ev)
        return out

    def process_shm(self, blobs,
                    compute_stream: Optional[torch.cuda.Stream] = None
                    ) -> Optional[torch.Tensor]:
        """Single-thread form: decode_shm + upload_shm."""
        decoded = self.decode_shm(blobs)
        if decoded is None:
            return None
        return self.upload_shm(decoded, len(blobs), compute_stream)

    def warm_pool(self):
        try:
            self._ensure_shm().warm()
        except Exception:
            pass  # /dev/shm unavailable: pipe path still works
        if not self._pool_warm:
            from PIL import Image
            buf = io.BytesIO()
            Image.new("RGB", (8, 8)).save(buf, "PNG")
            list(self._pool.map(_decode_worker,
                                [(buf.getvalue(), self.c)] * 8))
            self._pool_warm = True

    def process_arrays(self, arrs: Sequence[np.ndarray],
                       compute_stream: Optional[torch.cuda.Stream] = None
                       ) -> torch.Tensor:
        """decoded uint8 HWC arrays (any sizes) -> f32 NHWC batch."""
        n = len(arrs)
        assert n <= self.max_batch
        out = self._out[:n]
        # pack all images into one pinned arena (distinct offsets: safe
        # with the async H2D copies still in flight)
        total = sum(a.nbytes for a in arrs)
        arena = self._pin.get("arena")
        if arena is None or arena.numel() < total:
            arena = torch.empty(max(total, 1 << 22), dtype=torch.uint8,
                                pin_memory=True)
            self._pin["arena"] = arena
        meta = np.empty((n, 4), np.int32)
        off = 0
        for i, a in enumerate(arrs):
            arena[off:off + a.nbytes].copy_(torch.from_numpy(a.reshape(-1)))
            meta[i] = (off, a.shape[0], a.shape[1], a.shape[2])
            off += a.nbytes
        with torch.cuda.stream(self.stream):
            # ONE H2D + ONE kernel for the whole super-batch (a per-image
            # Python loop here gets GIL-starved under request load)
            arena_gpu = arena[:off].to(self.device, non_blocking=True)
            meta_gpu = torch.from_numpy(meta).to(self.device,
                                                 non_blocking=True)
            self.E.resize_norm_batch(arena_gpu, meta_gpu, out, self.norm)
        ev = torch.cuda.Event()
        ev.record(self.stream)
        target = compute_stream or torch.cuda.current_stream()
        target.wait_event(ev)
        return out[:n]

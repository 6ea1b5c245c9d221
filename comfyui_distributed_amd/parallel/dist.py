"""torch.distributed context: one process per GPU, RCCL over xGMI.

This replaces the reference's process-per-GPU HTTP mesh
(api/orchestration/dispatch.py + utils/network.py) for the intra-node data
plane: the control plane becomes a TCPStore (tile_queue.py), the data plane
NCCL-API collectives (= RCCL on ROCm). On CPU test hosts the same code runs
on the gloo backend.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist

from ..utils import constants
from ..utils.logging import log


@dataclass
class DistContext:
    rank: int
    world_size: int
    device: torch.device
    backend: str

    @property
    def is_master(self) -> bool:
        return self.rank == 0

    def barrier(self):
        if self.world_size > 1:
            dist.barrier()

    def sync_device(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)


def init_from_env(backend: str | None = None, timeout_s: float = 600.0) -> DistContext:
    """Initialize from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_*).

    Single-process (no env / WORLD_SIZE=1) returns a degenerate context
    without initializing a process group.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world <= 1:
        return DistContext(0, 1, device, "none")
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    os.environ.setdefault("MASTER_ADDR", constants.DEFAULT_MASTER_ADDR)
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    log(f"rank {rank}/{world} initialized ({backend}, {device})")
    return DistContext(rank, world, device, backend)


def gather_tensor_lists(
    ctx: DistContext,
    tensors: list[torch.Tensor],
    meta: list[tuple],
) -> tuple[list[torch.Tensor], list[tuple]] | None:
    """Gather variable-length tensor lists (+ metadata tuples) to rank 0.

    Each rank contributes N_i tensors of IDENTICAL shape (tile results).
    Returns (all_tensors, all_meta) on rank 0, None elsewhere. Uses
    all_gather for counts then batched p2p send/recv — on RCCL each source
    rank streams over its own xGMI link to rank 0 rather than serializing
    through a symmetric collective (SURVEY.md §5.8 topology note).
    """
    if ctx.world_size == 1:
        return tensors, meta
    device = ctx.device if ctx.backend == "nccl" else torch.device("cpu")
    count = torch.tensor([len(tensors)], dtype=torch.int64, device=device)
    counts = [torch.zeros_like(count) for _ in range(ctx.world_size)]
    dist.all_gather(counts, count)
    counts = [int(c.item()) for c in counts]

    # metadata: fixed-width int64 rows
    meta_width = len(meta[0]) if meta else 2
    if tensors:
        shape = tensors[0].shape
        payload = torch.stack([t.to(device) for t in tensors]).contiguous()
        meta_t = torch.tensor(meta, dtype=torch.int64, device=device)
    else:
        shape = None
        payload = None
        meta_t = None

    # shape+dtype agreement: broadcast from the first non-empty rank.
    # Payloads travel in their SOURCE dtype — bf16 tile results cost half
    # the xGMI bytes the old fp32 upcast did (r1 verdict item 7).
    dtype_codes = {torch.float32: 0, torch.bfloat16: 1, torch.float16: 2,
                   torch.uint8: 3, torch.int64: 4}
    code_dtypes = {v: k for k, v in dtype_codes.items()}
    shape_src = next((r for r, c in enumerate(counts) if c > 0), None)
    if shape_src is None:
        return ([], []) if ctx.is_master else None
    shape_t = torch.zeros(10, dtype=torch.int64, device=device)
    if ctx.rank == shape_src and shape is not None:
        dims = torch.tensor(shape, dtype=torch.int64, device=device)
        shape_t[0] = len(shape)
        shape_t[1 : 1 + len(shape)] = dims
        shape_t[9] = dtype_codes.get(payload.dtype, 0)
    dist.broadcast(shape_t, src=shape_src)
    ndim = int(shape_t[0].item())
    ref_shape = tuple(int(x) for x in shape_t[1 : 1 + ndim])
    wire_dtype = code_dtypes[int(shape_t[9].item())]

    # One batched p2p group: every source's isend pairs with a
    # pre-posted irecv on rank 0, so all sources stream CONCURRENTLY over
    # their own xGMI links instead of rank 0 draining them one at a time
    # (xGMI is 7 point-to-point links per GPU — serial recv leaves 6 idle).
    if ctx.is_master:
        bufs: dict[int, tuple[torch.Tensor, torch.Tensor]] = {}
        ops = []
        for src in range(1, ctx.world_size):
            n = counts[src]
            if n == 0:
                continue
            buf = torch.empty((n, *ref_shape), dtype=wire_dtype, device=device)
            mbuf = torch.empty((n, meta_width), dtype=torch.int64, device=device)
            bufs[src] = (buf, mbuf)
            ops.append(dist.P2POp(dist.irecv, buf, src))
            ops.append(dist.P2POp(dist.irecv, mbuf, src))
        if ops:
            for work in dist.batch_isend_irecv(ops):
                work.wait()
        all_tensors: list[torch.Tensor] = list(tensors)
        all_meta: list[tuple] = list(meta)
        for src in range(1, ctx.world_size):
            if src not in bufs:
                continue
            buf, mbuf = bufs[src]
            for i in range(counts[src]):
                all_tensors.append(buf[i])
                all_meta.append(tuple(int(x) for x in mbuf[i]))
        return all_tensors, all_meta
    if counts[ctx.rank] > 0:
        ops = [dist.P2POp(dist.isend, payload.to(wire_dtype), 0),
               dist.P2POp(dist.isend, meta_t, 0)]
        for work in dist.batch_isend_irecv(ops):
            work.wait()
    return None


_DTYPE_CODES = {
    "float32": torch.float32,
    "float16": torch.float16,
    "bfloat16": torch.bfloat16,
    "uint8": torch.uint8,
    "int64": torch.int64,
}


class ResultMailbox:
    """Streamed, fault-tolerant result transport: workers p2p-send each
    finished chunk to rank 0 as it completes (overlapping xGMI transfers
    with the next chunk's compute), in the tensors' SOURCE dtype (bf16 tile
    results cost half the link bytes of the old fp32 end-of-job gather).

    Coordination rides on the job KV store instead of collectives so a dead
    rank cannot hang the group: the sender posts its isends FIRST and only
    then publishes a chunk descriptor (count/shape/dtype + metadata) under
    ``mb:<job>:<rank>:<seq>``; rank 0 polls descriptors, posts the matching
    irecvs, and bounds each wait — a sender that died mid-transfer times
    out and is reported instead of deadlocking the job. Replaces the
    reference's submit_tiles HTTP POSTs + 5-retry backoff
    (upscale/worker_comms.py:16-108) for the intra-node path.
    """

    def __init__(self, ctx: DistContext, store, job_id: str,
                 recv_timeout: float = 30.0):
        self.ctx = ctx
        self.store = store
        self.job_id = job_id
        self.recv_timeout = recv_timeout
        self._seq = 0
        self._pending: list[tuple[list, torch.Tensor, torch.Tensor]] = []
        self._next_seq = dict.fromkeys(range(1, ctx.world_size), 0)
        self.failed_ranks: set[int] = set()

    def _k(self, rank: int, name) -> str:
        return f"mb:{self.job_id}:{rank}:{name}"

    def _read(self, key: str) -> bytes:
        # non-blocking store read (TCPStore.get blocks on missing keys)
        return self.store.compare_set(key, "", "")

    @property
    def _wire_device(self) -> torch.device:
        return self.ctx.device if self.ctx.backend == "nccl" else torch.device("cpu")

    # -- worker side --------------------------------------------------------

    def send_chunk(self, tensors: list[torch.Tensor], meta: list[tuple]) -> None:
        """Post async sends for one chunk of same-shape tensors and publish
        its descriptor. Returns immediately; call flush() before exiting."""
        if not tensors:
            return
        import json

        device = self._wire_device
        payload = torch.stack([t.to(device) for t in tensors]).contiguous()
        meta_t = torch.tensor(meta, dtype=torch.int64, device=device).contiguous()
        ops = [dist.P2POp(dist.isend, payload, 0),
               dist.P2POp(dist.isend, meta_t, 0)]
        works = dist.batch_isend_irecv(ops)
        # keep buffer refs alive until the transfer completes
        self._pending.append((works, payload, meta_t))
        desc = json.dumps({
            "n": len(tensors),
            "shape": list(payload.shape[1:]),
            "dtype": str(payload.dtype).replace("torch.", ""),
            "meta": [list(int(x) for x in m) for m in meta],
        })
        self.store.set(self._k(self.ctx.rank, self._seq), desc)
        self._seq += 1

    def flush(self) -> None:
        """Wait for all outstanding sends. The bound is deliberately
        generous: a send only completes once rank 0 posts its receive, and
        rank 0 may legitimately be inside a multi-minute first chunk
        (MIOpen warmup, graph capture) before it drains — a short bound
        here would crash healthy workers during warmup. A genuinely dead
        rank 0 ends the whole job regardless."""
        deadline = datetime.timedelta(
            seconds=max(self.recv_timeout * 10.0, 900.0))
        for works, _p, _m in self._pending:
            for w in works:
                w.wait(deadline)
        self._pending.clear()

    def finish(self) -> None:
        """Publish the final chunk count; rank 0 uses it for termination."""
        self.flush()
        self.store.set(self._k(self.ctx.rank, "done"), str(self._seq))

    # -- rank-0 side ---------------------------------------------------------

    def drain(self, block: bool = False) -> tuple[list[torch.Tensor], list[tuple]]:
        """Collect every chunk whose descriptor is visible. Returns
        (tensors, meta) accumulated across sources. A source whose transfer
        times out joins ``failed_ranks``; its lost tiles surface through the
        caller's missing-result takeover."""
        import json

        out_t: list[torch.Tensor] = []
        out_m: list[tuple] = []
        device = self._wire_device
        timeout = datetime.timedelta(
            seconds=self.recv_timeout if block else max(self.recv_timeout, 5.0))
        for src in sorted(self._next_seq):
            if src in self.failed_ranks:
                continue
            while True:
                raw = self._read(self._k(src, self._next_seq[src]))
                if not raw:
                    break
                d = json.loads(raw)
                buf = torch.empty((d["n"], *d["shape"]),
                                  dtype=_DTYPE_CODES[d["dtype"]], device=device)
                mwidth = len(d["meta"][0]) if d["meta"] else 2
                mbuf = torch.empty((d["n"], mwidth), dtype=torch.int64,
                                   device=device)
                ops = [dist.P2POp(dist.irecv, buf, src),
                       dist.P2POp(dist.irecv, mbuf, src)]
                try:
                    for w in dist.batch_isend_irecv(ops):
                        w.wait(timeout)
                except Exception:
                    log(f"mailbox: recv from rank {src} timed out — dropping rank")
                    self.failed_ranks.add(src)
                    break
                self._next_seq[src] += 1
                for i in range(d["n"]):
                    out_t.append(buf[i])
                    out_m.append(tuple(int(x) for x in mbuf[i]))
        return out_t, out_m

    def rank_finished(self, rank: int) -> bool:
        """True once ``rank`` published its done marker and every chunk it
        announced has been drained."""
        if rank in self.failed_ranks:
            return True
        raw = self._read(self._k(rank, "done"))
        if not raw:
            return False
        return self._next_seq[rank] >= int(raw)


def broadcast_tensor(ctx: DistContext, t: torch.Tensor | None, src: int = 0):
    """Broadcast a tensor (shape+dtype negotiated) from src to all ranks."""
    if ctx.world_size == 1:
        return t
    device = ctx.device if ctx.backend == "nccl" else torch.device("cpu")
    shape_t = torch.zeros(9, dtype=torch.int64, device=device)
    if ctx.rank == src:
        assert t is not None
        shape_t[0] = t.dim()
        for i, s in enumerate(t.shape):
            shape_t[1 + i] = s
    dist.broadcast(shape_t, src=src)
    ndim = int(shape_t[0].item())
    shape = tuple(int(x) for x in shape_t[1 : 1 + ndim])
    if ctx.rank == src:
        buf = t.to(device=device, dtype=torch.float32).contiguous()
    else:
        buf = torch.empty(shape, dtype=torch.float32, device=device)
    dist.broadcast(buf, src=src)
    return buf

"""Distributed runtime context: one process per GPU, torch.distributed over
RCCL (backend "nccl" on ROCm == RCCL over xGMI intra-node) or gloo for
CPU-only tests.

Reference analogue: the shuffle transport environment (GpuShuffleEnv /
RapidsShuffleInternalManagerBase). The MI355X-native design replaces the
UCX client/server with collective all-to-all over the fully-connected xGMI
mesh (SURVEY.md §5.8 MI355X mapping).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as td


class DistContext:
    def __init__(self):
        self.initialized = td.is_available() and td.is_initialized()
        self.rank = td.get_rank() if self.initialized else 0
        self.world = td.get_world_size() if self.initialized else 1
        self.backend = td.get_backend() if self.initialized else None

    @property
    def is_multi(self) -> bool:
        return self.initialized and self.world > 1


_ctx: Optional[DistContext] = None

# Host-path shuffle compression codec (reference analogue: the nvcomp /
# lz4 shuffle codecs — GpuCompressedColumnVector). Device buffers stay
# uncompressed: xGMI moves them at link speed and compressing would force
# a host round-trip; the codec applies to the gloo (host) transport only.
_codec: Optional[str] = None


_wave_bytes = 1 << 30


def set_wave_bytes(n: int):
    global _wave_bytes
    _wave_bytes = int(n)


def wave_bytes() -> int:
    return _wave_bytes


def set_codec(name: Optional[str]):
    global _codec
    _codec = None if not name or name == "none" else name


def _compress(t: torch.Tensor) -> torch.Tensor:
    if _codec is None or t.is_cuda or t.numel() == 0:
        return t
    import pyarrow as pa

    raw = t.numpy().tobytes()
    comp = pa.Codec(_codec).compress(raw, asbytes=True)
    import numpy as np

    hdr = np.array([len(raw)], dtype=np.int64).tobytes()
    return torch.frombuffer(bytearray(hdr + comp), dtype=torch.uint8)


def _decompress(t: torch.Tensor) -> torch.Tensor:
    if _codec is None or t.is_cuda or t.numel() == 0:
        return t
    import pyarrow as pa

    b = t.numpy().tobytes()
    raw_len = int.from_bytes(b[:8], "little")
    out = pa.Codec(_codec).decompress(b[8:], raw_len, asbytes=True)
    return torch.frombuffer(bytearray(out), dtype=torch.uint8)


def ctx() -> DistContext:
    global _ctx
    if _ctx is None or (_ctx.initialized != (td.is_available() and td.is_initialized())):
        _ctx = DistContext()
    return _ctx


def all_to_all_bytes(send: List[torch.Tensor]) -> List[torch.Tensor]:
    """Exchange one uint8 buffer per destination rank; returns one buffer per
    source rank. NCCL/RCCL path: all_to_all_single over xGMI (device
    buffers). Gloo path (CPU tests): batched isend/irecv."""
    c = ctx()
    world = c.world
    assert len(send) == world
    if c.backend != "nccl":
        send = [_compress(t) for t in send]
    sizes = torch.tensor([t.numel() for t in send], dtype=torch.int64)
    dev = send[0].device if send else torch.device("cpu")
    use_device = dev.type == "cuda"
    sizes_d = sizes.to(dev) if use_device else sizes
    recv_sizes = torch.empty_like(sizes_d)
    td.all_to_all_single(recv_sizes, sizes_d)
    recv_sizes = recv_sizes.cpu().tolist()
    send_sizes = sizes.tolist()

    if c.backend == "nccl":
        send_buf = torch.cat(send) if world > 1 else send[0]
        recv_buf = torch.empty(sum(recv_sizes), dtype=torch.uint8, device=dev)
        td.all_to_all_single(recv_buf, send_buf,
                             output_split_sizes=recv_sizes,
                             input_split_sizes=send_sizes)
        out, off = [], 0
        for s in recv_sizes:
            out.append(recv_buf[off:off + s])
            off += s
        return out

    # gloo: point-to-point
    recv = [torch.empty(recv_sizes[src], dtype=torch.uint8)
            for src in range(world)]
    ops = []
    for dst in range(world):
        if dst == c.rank:
            continue
        if send_sizes[dst]:
            ops.append(td.P2POp(td.isend, send[dst], dst))
        if recv_sizes[dst]:
            ops.append(td.P2POp(td.irecv, recv[dst], dst))
    recv[c.rank] = send[c.rank].clone()
    if ops:
        for w in td.batch_isend_irecv(ops):
            w.wait()
    return [_decompress(t) for t in recv]


def all_gather_bytes(buf: torch.Tensor) -> List[torch.Tensor]:
    """Gather one uint8 buffer from every rank (used to merge keyless /
    broadcast-size aggregates on all ranks)."""
    c = ctx()
    size = torch.tensor([buf.numel()], dtype=torch.int64)
    dev = buf.device
    use_device = dev.type == "cuda"
    sizes = [torch.zeros(1, dtype=torch.int64, device=dev if use_device else "cpu")
             for _ in range(c.world)]
    td.all_gather(sizes, size.to(dev) if use_device else size)
    sizes = [int(s.item()) for s in sizes]
    mx = max(sizes + [1])
    padded = torch.zeros(mx, dtype=torch.uint8, device=dev)
    padded[:buf.numel()] = buf
    outs = [torch.empty(mx, dtype=torch.uint8, device=dev)
            for _ in range(c.world)]
    td.all_gather(outs, padded)
    return [o[:s] for o, s in zip(outs, sizes)]

"""Transport interface: one-sided RDMA-write/read semantics.

A transport owns a *source* staging side and a *destination* region (the
"registered MR").  `post(i)` enqueues message i (size msg_bytes) toward
the destination at offset (i % msgs_per_region) * msg_bytes; `flush()`
waits for all outstanding messages — mirroring
ibv_post_send(IBV_WR_RDMA_WRITE) + CQ polling.  For `direction="read"`
the roles flip (region -> staging), mirroring RDMA READ.

`integrity_check()` proves the data path end to end: it moves the whole
region with deterministic per-message payloads and verifies the receiving
side (on-GPU via the CRC/verify kernels when the receiver is HBM) —
never inside a timed section.
"""
from __future__ import annotations

import abc


class Transport(abc.ABC):
    name = "base"

    def __init__(self, msg_bytes: int, region_bytes: int,
                 inflight: int | None = 8, direction: str = "write", **_):
        if region_bytes % msg_bytes:
            raise ValueError("region must be a multiple of msg size")
        if direction not in ("write", "read"):
            raise ValueError(direction)
        if not inflight:
            inflight = 8
        self.msg_bytes = msg_bytes
        self.region_bytes = region_bytes
        self.inflight = max(1, min(inflight, region_bytes // msg_bytes))
        self.direction = direction

    # -- data plane ----------------------------------------------------
    @abc.abstractmethod
    def post(self, i: int) -> None:
        """Enqueue message i (non-blocking where the backend allows)."""

    @abc.abstractmethod
    def flush(self) -> None:
        """Complete every outstanding message."""

    def post_many(self, start: int, n: int) -> None:
        """Enqueue messages start..start+n-1 (backends may vectorize the
        WQE writes; semantics identical to n post() calls)."""
        for i in range(start, start + n):
            self.post(i)

    # -- integrity plane (never timed) ---------------------------------
    @abc.abstractmethod
    def integrity_check(self, seed: int) -> int:
        """Move the whole region with pattern payloads; return mismatching
        8-byte words at the receiver (0 = intact)."""

    def close(self) -> None:  # optional
        pass

    @property
    def msgs_per_region(self) -> int:
        return self.region_bytes // self.msg_bytes

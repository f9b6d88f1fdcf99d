"""CPU loopback transport: BASELINE config 1 ("host-malloc ibv_reg_mr +
ib_write_bw loopback") without hardware.  Used by the CPU test tier and
as the bench fallback where no GPU/HCA exists."""
from __future__ import annotations

import numpy as np

from ..utils import pattern
from .base import Transport


class FakeTransport(Transport):
    name = "fake"

    def __init__(self, msg_bytes: int, region_bytes: int, **kw):
        super().__init__(msg_bytes, region_bytes, **kw)
        self.staging = np.zeros((self.inflight, msg_bytes), dtype=np.uint8)
        self.region = np.zeros(region_bytes, dtype=np.uint8)

    def post(self, i: int) -> None:
        slot = self.staging[i % self.inflight]
        off = (i % self.msgs_per_region) * self.msg_bytes
        dst = self.region[off : off + self.msg_bytes]
        if self.direction == "write":
            dst[:] = slot
        else:
            slot[:] = dst

    def flush(self) -> None:  # synchronous backend
        pass

    def integrity_check(self, seed: int) -> int:
        ref = pattern.fill_reference(self.region_bytes, seed)
        if self.direction == "write":
            for i in range(self.msgs_per_region):
                off = i * self.msg_bytes
                self.staging[i % self.inflight][:] = ref[off : off + self.msg_bytes]
                self.post(i)
            self.flush()
            got = self.region
            return int(
                np.count_nonzero(
                    got.view(np.uint64) != ref.view(np.uint64)))
        # read: pattern the region, pull into staging slot by slot
        self.region[:] = ref
        bad = 0
        for i in range(self.msgs_per_region):
            self.post(i)
            self.flush()
            off = i * self.msg_bytes
            bad += int(
                np.count_nonzero(
                    self.staging[i % self.inflight].view(np.uint64)
                    != ref[off : off + self.msg_bytes].view(np.uint64)))
        return bad

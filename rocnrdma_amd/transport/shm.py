"""Two-process one-sided transport over POSIX shared memory.

The loopback transports merge both RDMA endpoints into one process;
this one separates the roles the way ib_write_bw does:

- the TARGET registers a memory region and sits passive (it only
  answers the out-of-band bootstrap and the final verification request
  — zero involvement per message, like a remote HCA's memory);
- the INITIATOR attaches and performs one-sided writes/reads at
  arbitrary offsets.

Data plane is mmap'd shared memory (host-side config-1 analog); the
OOB bootstrap (rocnrdma_amd/transport/oob.py) is the same exchange a
remote verbs deployment performs with GID/QPN/rkey instead of an shm
name.  Remote verification: the initiator sends {seed}; the TARGET
verifies its own region against the pattern and reports mismatches —
end-to-end proof that one-sided writes landed in the *other process's*
registered memory.
"""
from __future__ import annotations

from multiprocessing import shared_memory

import numpy as np

from ..utils import pattern
from .base import Transport
from .oob import OobClient, OobServer


def target_serve(port_conn, region_bytes: int, port: int = 0) -> None:
    """Run the passive target (in a child process).  port_conn is a
    multiprocessing Pipe end (or None) used to report the OOB port."""
    shm = shared_memory.SharedMemory(create=True, size=region_bytes)
    server = OobServer(port=port)
    if port_conn is not None:
        port_conn.send(server.port)
    else:
        print(f"shm target: region {region_bytes} bytes, OOB port "
              f"{server.port}", flush=True)
    try:
        server.accept()
        server.send({"shm_name": shm.name, "region_bytes": region_bytes})
        # passive until the initiator asks for verification or bye;
        # an initiator crash (socket close) is a normal shutdown
        while True:
            try:
                msg = server.recv()
            except (ConnectionError, OSError):
                break
            if msg.get("op") == "verify":
                ref = pattern.fill_reference(region_bytes, msg["seed"])
                got = np.frombuffer(shm.buf, dtype=np.uint8)
                bad = int(np.count_nonzero(
                    got.view(np.uint64) != ref.view(np.uint64)))
                server.send({"bad": bad})
            elif msg.get("op") == "fill":
                # for read-direction tests: target patterns its region
                ref = pattern.fill_reference(region_bytes, msg["seed"])
                np.frombuffer(shm.buf, dtype=np.uint8)[:] = ref
                server.send({"ok": 1})
            else:
                break
    finally:
        server.close()
        shm.close()
        shm.unlink()


class ShmInitiatorTransport(Transport):
    """Initiator: one-sided writes/reads into the target's region."""

    name = "shm"

    def __init__(self, msg_bytes: int, region_bytes: int, host: str,
                 port: int, **kw):
        super().__init__(msg_bytes, region_bytes, **kw)
        self.oob = OobClient(host, port)
        info = self.oob.recv()
        if info["region_bytes"] != region_bytes:
            raise ValueError("target region size mismatch")
        self.shm = shared_memory.SharedMemory(name=info["shm_name"])
        self.region = np.frombuffer(self.shm.buf, dtype=np.uint8)
        self.staging = np.zeros((self.inflight, msg_bytes), dtype=np.uint8)

    def post(self, i: int) -> None:
        slot = self.staging[i % self.inflight]
        off = (i % self.msgs_per_region) * self.msg_bytes
        dst = self.region[off:off + self.msg_bytes]
        if self.direction == "write":
            dst[:] = slot          # one-sided: target not involved
        else:
            slot[:] = dst

    def flush(self) -> None:
        pass

    def integrity_check(self, seed: int) -> int:
        if self.direction == "write":
            ref = pattern.fill_reference(self.region_bytes, seed)
            for i in range(self.msgs_per_region):
                off = i * self.msg_bytes
                self.staging[i % self.inflight][:] = \
                    ref[off:off + self.msg_bytes]
                self.post(i)
            self.flush()
            # REMOTE verification: the target checks its own memory
            self.oob.send({"op": "verify", "seed": seed})
            return int(self.oob.recv()["bad"])
        # read: target patterns its region; we pull and verify locally
        self.oob.send({"op": "fill", "seed": seed})
        self.oob.recv()
        ref = pattern.fill_reference(self.region_bytes, seed)
        bad = 0
        for i in range(self.msgs_per_region):
            self.post(i)
            off = i * self.msg_bytes
            got = self.staging[i % self.inflight]
            bad += int(np.count_nonzero(
                got.view(np.uint64)
                != ref[off:off + self.msg_bytes].view(np.uint64)))
        return bad

    def close(self) -> None:
        try:
            self.oob.send({"op": "bye"})
        except OSError:
            pass
        self.oob.close()
        # release the numpy views before closing the mapping
        self.region = None
        self.shm.close()

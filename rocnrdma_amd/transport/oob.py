"""Out-of-band bootstrap for two-process transports.

perftest-style rendezvous: target listens, initiator connects, both
exchange one JSON object (region geometry, shm name / in verbs terms
GID+QPN+rkey), then reuse the socket for end-of-run control messages.
Shared by the shm transport (tested here) and by a remote verbs
deployment (same exchange with verbs fields, HCA hosts).
"""
from __future__ import annotations

import json
import socket


def _send_obj(sock: socket.socket, obj: dict) -> None:
    data = json.dumps(obj).encode() + b"\n"
    sock.sendall(data)


def _recv_obj(sock_file) -> dict:
    line = sock_file.readline()
    if not line:
        raise ConnectionError("OOB peer closed")
    return json.loads(line)


class OobServer:
    """Target side: bind, accept one initiator, exchange objects."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.lsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.lsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.lsock.bind((host, port))
        self.lsock.listen(1)
        self.port = self.lsock.getsockname()[1]
        self.conn = None
        self._file = None

    def accept(self, timeout: float = 30.0):
        self.lsock.settimeout(timeout)
        self.conn, _ = self.lsock.accept()
        self._file = self.conn.makefile("r")

    def send(self, obj: dict) -> None:
        _send_obj(self.conn, obj)

    def recv(self, timeout: float = 60.0) -> dict:
        self.conn.settimeout(timeout)
        return _recv_obj(self._file)

    def close(self) -> None:
        for s in (self.conn, self.lsock):
            if s is not None:
                s.close()


class OobClient:
    """Initiator side."""

    def __init__(self, host: str, port: int, timeout: float = 30.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self._file = self.sock.makefile("r")

    def send(self, obj: dict) -> None:
        _send_obj(self.sock, obj)

    def recv(self, timeout: float = 60.0) -> dict:
        self.sock.settimeout(timeout)
        return _recv_obj(self._file)

    def close(self) -> None:
        self.sock.close()

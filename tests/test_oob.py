"""OOB bootstrap protocol tests (the exchange both the shm transport
and the remote verbs deployment ride on)."""
import threading

import pytest

from rocnrdma_amd.transport.oob import OobClient, OobServer


def test_exchange_roundtrip():
    server = OobServer()
    result = {}

    def serve():
        server.accept()
        msg = server.recv()
        result["got"] = msg
        server.send({"reply": msg["x"] * 2, "extra": [1, 2]})

    th = threading.Thread(target=serve, daemon=True)
    th.start()
    cli = OobClient("127.0.0.1", server.port)
    cli.send({"x": 21, "name": "init"})
    reply = cli.recv()
    th.join(timeout=10)
    assert result["got"] == {"x": 21, "name": "init"}
    assert reply == {"reply": 42, "extra": [1, 2]}
    cli.close()
    server.close()


def test_peer_close_raises():
    server = OobServer()

    def serve():
        server.accept()
        server.conn.close()

    th = threading.Thread(target=serve, daemon=True)
    th.start()
    cli = OobClient("127.0.0.1", server.port)
    th.join(timeout=10)
    with pytest.raises((ConnectionError, OSError)):
        cli.recv(timeout=5)
    cli.close()
    server.close()


def test_multiple_messages_in_order():
    server = OobServer()

    def serve():
        server.accept()
        for i in range(5):
            server.send({"seq": i})

    th = threading.Thread(target=serve, daemon=True)
    th.start()
    cli = OobClient("127.0.0.1", server.port)
    for i in range(5):
        assert cli.recv()["seq"] == i
    th.join(timeout=10)
    cli.close()
    server.close()

"""TCP bus bridge: the wire seam for EXTERNAL CAP workers.

The reference's workers attach over NATS (`sdk/runtime/worker.go:50-148`,
subjects in capsdk/constants.go); this node's bus is in-process, so the
bridge exposes a NATS-subset wire protocol carrying CAP v2 `BusPacket`
frames over TCP. Byte compatibility of the payloads is the vendored
`protocol/capv2.proto` contract (proven against the google.protobuf runtime
in tests/test_capv2_interop.py) — any protoc-generated CAP binding can
speak this framing.

Frame format (all integers big-endian u32):

    [op u8] [subject_len u32] [subject utf-8] [payload_len u32] [payload]

ops:
    0x01 PUB   payload = BusPacket bytes; published onto the node bus
    0x02 SUB   subject = pattern; payload = queue-group name (may be empty)
    0x03 MSG   server -> client delivery (payload = BusPacket bytes)
    0x04 PING / 0x05 PONG (liveness)

Semantics mirror the in-process bus (and NATS): `*`/`>` wildcards, queue
groups load-balance across subscribers (including in-process ones — an
external SUB joins the same group namespace), publishes from the socket get
the same msg-id dedup/at-least-once treatment as local publishes.

Context/result transport stays on the compat HTTP surface, exactly like the
reference's Redis pointers: workers GET /api/v1/memory?ptr=<context_ptr>
and POST /api/v1/artifacts for the result blob, then publish a JobResult
with that pointer.
"""
from __future__ import annotations

import socket
import struct
import threading
from typing import List, Optional, Tuple

from ..protocol.capv2 import BusPacket

OP_PUB = 0x01
OP_SUB = 0x02
OP_MSG = 0x03
OP_PING = 0x04
OP_PONG = 0x05

MAX_FRAME = 8 << 20


def write_frame(sock: socket.socket, op: int, subject: str, payload: bytes) -> None:
    sub = subject.encode("utf-8")
    sock.sendall(struct.pack(">BI", op, len(sub)) + sub +
                 struct.pack(">I", len(payload)) + payload)


def read_frame(sock: socket.socket) -> Optional[Tuple[int, str, bytes]]:
    def read_exact(n: int) -> Optional[bytes]:
        buf = b""
        while len(buf) < n:
            chunk = sock.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf

    hdr = read_exact(5)
    if hdr is None:
        return None
    op, slen = struct.unpack(">BI", hdr)
    if slen > MAX_FRAME:
        raise ValueError("subject too long")
    sub = read_exact(slen)
    if sub is None:
        return None
    plen_raw = read_exact(4)
    if plen_raw is None:
        return None
    (plen,) = struct.unpack(">I", plen_raw)
    if plen > MAX_FRAME:
        raise ValueError("frame too large")
    payload = read_exact(plen)
    if payload is None:
        return None
    return op, sub.decode("utf-8"), payload


class _BridgeConn:
    def __init__(self, server: "BusBridgeServer", sock: socket.socket):
        self.server = server
        self.sock = sock
        self.send_mu = threading.Lock()
        self.subs: List = []
        self.alive = True

    def deliver(self, subject: str, pkt: BusPacket) -> None:
        if not self.alive:
            raise ConnectionError("bridge connection closed")
        try:
            with self.send_mu:
                write_frame(self.sock, OP_MSG, subject, pkt.encode())
        except OSError:
            self.close()
            raise

    def close(self) -> None:
        if not self.alive:
            return
        self.alive = False
        for s in self.subs:
            s.unsubscribe()
        try:
            self.sock.close()
        except OSError:
            pass

    def serve(self) -> None:
        try:
            while self.alive:
                frame = read_frame(self.sock)
                if frame is None:
                    break
                op, subject, payload = frame
                if op == OP_PUB:
                    pkt = BusPacket.decode(payload)
                    self.server.node.bus.publish(subject, pkt)
                    # external publishes drive the node forward like HTTP
                    # submissions do (the reference's NATS consumers run in
                    # their own processes; here the tick loop is the pump)
                    self.server.node.drain()
                elif op == OP_SUB:
                    group = payload.decode("utf-8") or None
                    sub = self.server.node.bus.subscribe(
                        subject, self._make_handler(), queue_group=group)
                    self.subs.append(sub)
                elif op == OP_PING:
                    with self.send_mu:
                        write_frame(self.sock, OP_PONG, "", b"")
        except (OSError, ValueError):
            pass
        finally:
            self.close()

    def _make_handler(self):
        from .bus import RetryAfter

        def handler(subject: str, pkt: BusPacket) -> None:
            try:
                self.deliver(subject, pkt)
            except (ConnectionError, OSError) as e:
                # a dead wire client must never fail the PUBLISHER (the
                # scheduler's dispatch, a gateway submit). The subscription
                # is closed by deliver(); NAK so durable subjects redeliver
                # to a surviving queue-group member, exactly like a NATS
                # client vanishing mid-message (bus/nats.go:146-168) — and the
                # staleness reconciler covers a direct-subject worker that
                # died holding a job
                raise RetryAfter(0.05, cause=f"bridge client gone: {e}")

        return handler


class BusBridgeServer:
    """Accepts external bus clients for a running Node."""

    def __init__(self, node, host: str = "127.0.0.1", port: int = 0):
        self.node = node
        self._lsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._lsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._lsock.bind((host, port))
        self._lsock.listen(16)
        self.port = self._lsock.getsockname()[1]
        self._conns: List[_BridgeConn] = []
        self._accept_thread: Optional[threading.Thread] = None
        self._stopping = False

    def start(self) -> "BusBridgeServer":
        self._accept_thread = threading.Thread(target=self._accept_loop, daemon=True)
        self._accept_thread.start()
        return self

    def _accept_loop(self) -> None:
        while not self._stopping:
            try:
                sock, _ = self._lsock.accept()
            except OSError:
                return
            conn = _BridgeConn(self, sock)
            self._conns = [c for c in self._conns if c.alive]
            self._conns.append(conn)
            threading.Thread(target=conn.serve, daemon=True).start()

    def stop(self) -> None:
        self._stopping = True
        try:
            self._lsock.close()
        except OSError:
            pass
        for c in list(self._conns):
            c.close()


class BridgeClient:
    """Client side of the bridge (what an external worker runtime uses)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 4230, timeout: float = 10.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.sock.settimeout(None)
        self._send_mu = threading.Lock()

    def publish(self, subject: str, pkt: BusPacket) -> None:
        with self._send_mu:
            write_frame(self.sock, OP_PUB, subject, pkt.encode())

    def subscribe(self, pattern: str, queue_group: str = "") -> None:
        with self._send_mu:
            write_frame(self.sock, OP_SUB, pattern, queue_group.encode("utf-8"))

    def next_message(self) -> Optional[Tuple[str, BusPacket]]:
        """Blocking read of the next delivered message (None on close)."""
        while True:
            frame = read_frame(self.sock)
            if frame is None:
                return None
            op, subject, payload = frame
            if op == OP_MSG:
                return subject, BusPacket.decode(payload)
            if op == OP_PONG:
                continue

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass

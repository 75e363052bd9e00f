"""Minimal dependency-free RFC 6455 WebSocket server.

The image has no `websockets`/`wsproto` package, so MegaScope's servers
speak the protocol directly over sockets: HTTP upgrade handshake, masked
client frames, text/ping/pong/close opcodes, no extensions.  One client
at a time (the visualization frontend), which matches the reference's
usage of websockets.sync.
"""

from __future__ import annotations

import base64
import hashlib
import json
import socket
import struct
import threading
from typing import Callable, Optional

_GUID = "258EAFA5-E914-47DA-95CA-C5AB0DC85B11"

OP_TEXT = 0x1
OP_BIN = 0x2
OP_CLOSE = 0x8
OP_PING = 0x9
OP_PONG = 0xA


class WebSocketConnection:
    def __init__(self, sock: socket.socket, leftover: bytes = b""):
        self.sock = sock
        self._send_lock = threading.Lock()
        self.open = True
        self._buf = leftover  # bytes read past the HTTP handshake

    # ---------------------------------------------------------------- frames
    def _recv_exact(self, n: int) -> bytes:
        buf = b""
        if self._buf:
            take = min(n, len(self._buf))
            buf, self._buf = self._buf[:take], self._buf[take:]
        while len(buf) < n:
            chunk = self.sock.recv(n - len(buf))
            if not chunk:
                raise ConnectionError("socket closed")
            buf += chunk
        return buf

    def recv_message(self) -> Optional[str]:
        """Blocks until a full text message (handles ping/close)."""
        payload = b""
        while True:
            hdr = self._recv_exact(2)
            fin = hdr[0] & 0x80
            opcode = hdr[0] & 0x0F
            masked = hdr[1] & 0x80
            length = hdr[1] & 0x7F
            if length == 126:
                length = struct.unpack(">H", self._recv_exact(2))[0]
            elif length == 127:
                length = struct.unpack(">Q", self._recv_exact(8))[0]
            mask = self._recv_exact(4) if masked else None
            data = self._recv_exact(length) if length else b""
            if mask:
                data = bytes(b ^ mask[i % 4] for i, b in enumerate(data))
            if opcode == OP_CLOSE:
                self.open = False
                try:
                    self._send_frame(OP_CLOSE, b"")
                except OSError:
                    pass
                return None
            if opcode == OP_PING:
                self._send_frame(OP_PONG, data)
                continue
            if opcode == OP_PONG:
                continue
            payload += data
            if fin:
                return payload.decode("utf-8", errors="replace")

    def _send_frame(self, opcode: int, data: bytes):
        with self._send_lock:
            header = bytes([0x80 | opcode])
            n = len(data)
            if n < 126:
                header += bytes([n])
            elif n < (1 << 16):
                header += bytes([126]) + struct.pack(">H", n)
            else:
                header += bytes([127]) + struct.pack(">Q", n)
            self.sock.sendall(header + data)

    def send(self, message) -> None:
        if isinstance(message, (dict, list)):
            message = json.dumps(message)
        self._send_frame(OP_TEXT, message.encode("utf-8"))

    def close(self):
        self.open = False
        try:
            self._send_frame(OP_CLOSE, b"")
        except OSError:
            pass
        try:
            self.sock.close()
        except OSError:
            pass


class WebSocketServer:
    """serve(handler): accepts clients sequentially; handler(conn) runs
    until the client disconnects."""

    def __init__(self, host: str = "0.0.0.0", port: int = 5000):
        self.host = host
        self.port = port
        self._server_sock: Optional[socket.socket] = None
        self._stop = threading.Event()

    def _handshake(self, conn: socket.socket):
        data = b""
        while b"\r\n\r\n" not in data:
            chunk = conn.recv(4096)
            if not chunk:
                return None
            data += chunk
        headers = {}
        for line in data.split(b"\r\n")[1:]:
            if b":" in line:
                k, v = line.split(b":", 1)
                headers[k.strip().lower().decode()] = v.strip().decode()
        key = headers.get("sec-websocket-key")
        if key is None:
            conn.sendall(b"HTTP/1.1 400 Bad Request\r\n\r\n")
            return None
        accept = base64.b64encode(
            hashlib.sha1((key + _GUID).encode()).digest()).decode()
        conn.sendall((
            "HTTP/1.1 101 Switching Protocols\r\n"
            "Upgrade: websocket\r\n"
            "Connection: Upgrade\r\n"
            f"Sec-WebSocket-Accept: {accept}\r\n\r\n").encode())
        return data.partition(b"\r\n\r\n")[2]

    def serve_forever(self, handler: Callable[[WebSocketConnection], None]):
        self._server_sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._server_sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._server_sock.bind((self.host, self.port))
        self._server_sock.listen(2)
        self._server_sock.settimeout(1.0)
        while not self._stop.is_set():
            try:
                conn, _addr = self._server_sock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            try:
                leftover = self._handshake(conn)
                if leftover is not None:
                    handler(WebSocketConnection(conn, leftover))
            except (ConnectionError, OSError):
                pass
            finally:
                try:
                    conn.close()
                except OSError:
                    pass
        self._server_sock.close()

    def start_in_thread(self, handler) -> threading.Thread:
        t = threading.Thread(target=self.serve_forever, args=(handler,),
                             daemon=True)
        t.start()
        return t

    def stop(self):
        self._stop.set()


def ws_connect(host: str, port: int, timeout: float = 10.0
               ) -> WebSocketConnection:
    """Tiny client for tests: performs the upgrade and returns a connection."""
    sock = socket.create_connection((host, port), timeout=timeout)
    key = base64.b64encode(b"0123456789abcdef").decode()
    sock.sendall((
        f"GET / HTTP/1.1\r\nHost: {host}:{port}\r\n"
        "Upgrade: websocket\r\nConnection: Upgrade\r\n"
        f"Sec-WebSocket-Key: {key}\r\nSec-WebSocket-Version: 13\r\n\r\n"
    ).encode())
    data = b""
    while b"\r\n\r\n" not in data:
        data += sock.recv(4096)
    assert b"101" in data.split(b"\r\n")[0], data
    _, _, leftover = data.partition(b"\r\n\r\n")
    conn = WebSocketConnection(sock, leftover)

    # client frames must be masked; patch the sender
    def _send_frame_masked(opcode: int, payload: bytes, _conn=conn):
        import os as _os
        with _conn._send_lock:
            header = bytes([0x80 | opcode])
            n = len(payload)
            if n < 126:
                header += bytes([0x80 | n])
            elif n < (1 << 16):
                header += bytes([0x80 | 126]) + struct.pack(">H", n)
            else:
                header += bytes([0x80 | 127]) + struct.pack(">Q", n)
            mask = _os.urandom(4)
            masked = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
            _conn.sock.sendall(header + mask + masked)

    conn._send_frame = _send_frame_masked
    return conn

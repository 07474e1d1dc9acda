"""Minimal RFC 6455 WebSocket client (plain socket, no dependencies).

Used by core/browser.py to speak the Chrome DevTools Protocol — this image
has no websocket client library, and the framing needed for a local CDP
connection is small: client handshake, masked text frames out, server
frames in (text/binary/ping/close), 7/16/64-bit payload lengths.
"""
from __future__ import annotations

import base64
import hashlib
import os
import socket
import struct
from urllib.parse import urlparse

_GUID = "258EAFA5-E914-47DA-95CA-C5AB0DC85B11"


class WsClient:
    def __init__(self, url: str, timeout: float = 30.0):
        u = urlparse(url)
        if u.scheme != "ws":
            raise ValueError(f"only ws:// supported, got {url}")
        self.sock = socket.create_connection((u.hostname, u.port or 80),
                                             timeout=timeout)
        key = base64.b64encode(os.urandom(16)).decode()
        path = u.path + (f"?{u.query}" if u.query else "")
        req = (f"GET {path} HTTP/1.1\r\n"
               f"Host: {u.hostname}:{u.port or 80}\r\n"
               "Upgrade: websocket\r\nConnection: Upgrade\r\n"
               f"Sec-WebSocket-Key: {key}\r\n"
               "Sec-WebSocket-Version: 13\r\n\r\n")
        self.sock.sendall(req.encode())
        resp = b""
        while b"\r\n\r\n" not in resp:
            chunk = self.sock.recv(4096)
            if not chunk:
                raise ConnectionError("websocket handshake failed (closed)")
            resp += chunk
        head, _, rest = resp.partition(b"\r\n\r\n")
        if b" 101 " not in head.split(b"\r\n", 1)[0]:
            raise ConnectionError(f"websocket handshake rejected: "
                                  f"{head.splitlines()[0]!r}")
        expect = base64.b64encode(
            hashlib.sha1((key + _GUID).encode()).digest()).decode()
        if expect.encode() not in head:
            raise ConnectionError("websocket accept key mismatch")
        self._buf = rest

    # ------------------------------------------------------------- frames

    def send_text(self, text: str) -> None:
        payload = text.encode()
        mask = os.urandom(4)
        n = len(payload)
        if n < 126:
            header = struct.pack("!BB", 0x81, 0x80 | n)
        elif n < 1 << 16:
            header = struct.pack("!BBH", 0x81, 0x80 | 126, n)
        else:
            header = struct.pack("!BBQ", 0x81, 0x80 | 127, n)
        masked = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
        self.sock.sendall(header + mask + masked)

    def _read_exact(self, n: int, timeout: float) -> bytes | None:
        self.sock.settimeout(max(timeout, 0.01))
        while len(self._buf) < n:
            try:
                chunk = self.sock.recv(65536)
            except socket.timeout:
                return None
            if not chunk:
                raise ConnectionError("websocket closed")
            self._buf += chunk
        out, self._buf = self._buf[:n], self._buf[n:]
        return out

    def recv_text(self, timeout: float = 30.0) -> str | None:
        """Next text message (reassembling fragments); None on timeout.
        Pings are answered transparently."""
        message = b""
        while True:
            head = self._read_exact(2, timeout)
            if head is None:
                return None
            fin = head[0] & 0x80
            opcode = head[0] & 0x0F
            ln = head[1] & 0x7F
            if ln == 126:
                ln = struct.unpack("!H", self._read_exact(2, timeout))[0]
            elif ln == 127:
                ln = struct.unpack("!Q", self._read_exact(8, timeout))[0]
            if head[1] & 0x80:  # masked server frame (nonstandard): unmask
                mask = self._read_exact(4, timeout)
                data = self._read_exact(ln, timeout)
                data = bytes(b ^ mask[i % 4] for i, b in enumerate(data))
            else:
                data = self._read_exact(ln, timeout) if ln else b""
            if data is None:
                return None
            if opcode == 0x9:      # ping → pong
                mask = os.urandom(4)
                hdr = struct.pack("!BB", 0x8A, 0x80 | len(data))
                self.sock.sendall(hdr + mask + bytes(
                    b ^ mask[i % 4] for i, b in enumerate(data)))
                continue
            if opcode == 0x8:      # close
                raise ConnectionError("websocket closed by peer")
            if opcode in (0x1, 0x2, 0x0):
                message += data
                if fin:
                    return message.decode("utf-8", "replace")

    def close(self) -> None:
        try:
            mask = os.urandom(4)
            self.sock.sendall(struct.pack("!BB", 0x88, 0x80) + mask)
        except OSError:
            pass
        try:
            self.sock.close()
        except OSError:
            pass

"""Browser-automation layer (core/browser.py + utils/ws_client.py):
degradation contract without a chromium binary (this image has none) and
the WebSocket client's frame codec against a local echo server."""
import json
import socket
import struct
import threading

from room_amd.core import browser as br
from room_amd.core.web_tools import browser_action, close_browser


def test_no_chromium_degrades_like_reference(monkeypatch):
    monkeypatch.setattr(br, "find_chromium", lambda: None)
    out = br.browser_action("https://example.com", [{"type": "click"}])
    assert out["sessionId"] == "" and "not installed" in out["snapshot"]


def test_web_tools_fallback_to_fetch(monkeypatch):
    monkeypatch.setattr(br, "find_chromium", lambda: None)
    out = browser_action("s1", "click", selector="#x")
    assert out["ok"] is False and "browser runtime" in out["error"]
    # goto falls back to the urllib fetch path (which itself degrades
    # offline with ok=False + explicit error, never raises)
    out2 = browser_action("s1", "goto", url="https://example.com")
    assert "ok" in out2
    assert close_browser("s1") in (True, False)


def test_session_gc(monkeypatch):
    class FakeSession:
        def __init__(self):
            self.last_used = 0.0  # epoch → ancient
            self.closed = False
        def close(self):
            self.closed = True
    s = FakeSession()
    br._sessions["old"] = s
    br._gc_sessions()
    assert "old" not in br._sessions and s.closed


# ------------------------------------------------------- ws client codec

def _echo_ws_server(port_holder, stop):
    """Single-connection RFC6455 echo server (enough for the codec test)."""
    import base64, hashlib
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port_holder.append(srv.getsockname()[1])
    conn, _ = srv.accept()
    data = b""
    while b"\r\n\r\n" not in data:
        data += conn.recv(4096)
    key = [l.split(b": ")[1] for l in data.split(b"\r\n")
           if l.lower().startswith(b"sec-websocket-key")][0].decode()
    accept = base64.b64encode(hashlib.sha1(
        (key + "258EAFA5-E914-47DA-95CA-C5AB0DC85B11").encode()).digest()).decode()
    conn.sendall((f"HTTP/1.1 101 Switching Protocols\r\nUpgrade: websocket\r\n"
                  f"Connection: Upgrade\r\nSec-WebSocket-Accept: {accept}"
                  "\r\n\r\n").encode())
    while not stop.is_set():
        head = conn.recv(2)
        if len(head) < 2:
            break
        ln = head[1] & 0x7F
        if ln == 126:
            ln = struct.unpack("!H", conn.recv(2))[0]
        elif ln == 127:
            ln = struct.unpack("!Q", conn.recv(8))[0]
        mask = conn.recv(4)
        payload = b""
        while len(payload) < ln:
            payload += conn.recv(ln - len(payload))
        if (head[0] & 0x0F) == 0x8:
            break
        clear = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
        # echo back unmasked (server frames are unmasked)
        if len(clear) < 126:
            hdr = struct.pack("!BB", 0x81, len(clear))
        elif len(clear) < 1 << 16:
            hdr = struct.pack("!BBH", 0x81, 126, len(clear))
        else:
            hdr = struct.pack("!BBQ", 0x81, 127, len(clear))
        conn.sendall(hdr + clear)
    conn.close()
    srv.close()


def test_ws_client_roundtrip_small_and_large():
    from room_amd.utils.ws_client import WsClient
    port_holder, stop = [], threading.Event()
    t = threading.Thread(target=_echo_ws_server, args=(port_holder, stop),
                         daemon=True)
    t.start()
    while not port_holder:
        pass
    ws = WsClient(f"ws://127.0.0.1:{port_holder[0]}/cdp")
    ws.send_text(json.dumps({"id": 1, "method": "Page.enable"}))
    assert json.loads(ws.recv_text(timeout=5))["method"] == "Page.enable"
    big = "x" * 70_000  # exercises the 64-bit length path both ways
    ws.send_text(big)
    assert ws.recv_text(timeout=5) == big
    stop.set()
    ws.close()

"""Raw-socket HTTP/2 client for wire-conformance and fuzz testing.

Unlike the grpc C-core python client (which negotiates one canonical frame
pattern), this client hand-builds every frame, so tests can reproduce the
exact byte sequences other HTTP/2 stacks emit — in particular grpc-go's
(see goclient.py) — and deliberately malformed ones.  Response header
blocks are decoded with the native _h2tool HPACK inflater (nghttp2's), with
dynamic-table state persisting across a connection like a real peer's.

HPACK request encoding modes:
  - "plain":  literal-without-indexing, raw strings (RFC 7541 §6.2.2)
  - "never":  literal-never-indexed (§6.2.3)
  - "index":  nghttp2's deflater (incremental indexing + huffman — the
              shape grpc-go's hpack encoder produces)
"""

from __future__ import annotations

import socket
import struct
import time
from typing import Dict, List, Optional, Tuple

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"

DATA, HEADERS, PRIORITY, RST_STREAM, SETTINGS = 0, 1, 2, 3, 4
PUSH_PROMISE, PING, GOAWAY, WINDOW_UPDATE, CONTINUATION = 5, 6, 7, 8, 9

FLAG_END_STREAM = 0x1
FLAG_ACK = 0x1
FLAG_END_HEADERS = 0x4
FLAG_PADDED = 0x8


def frame(ftype: int, flags: int, stream_id: int, payload: bytes) -> bytes:
    return (
        struct.pack("!I", len(payload))[1:]
        + bytes([ftype, flags])
        + struct.pack("!I", stream_id & 0x7FFFFFFF)
        + payload
    )


def _hpack_int(value: int, prefix_bits: int, first_byte: int) -> bytes:
    """RFC 7541 §5.1 integer encoding under a first-byte pattern."""
    limit = (1 << prefix_bits) - 1
    if value < limit:
        return bytes([first_byte | value])
    out = bytearray([first_byte | limit])
    value -= limit
    while value >= 128:
        out.append((value & 0x7F) | 0x80)
        value >>= 7
    out.append(value)
    return bytes(out)


def hpack_plain(headers: List[Tuple[str, str]], never: bool = False) -> bytes:
    """Hand-rolled literal encoding, no huffman, no table use."""
    first = 0x10 if never else 0x00
    prefix = 4
    out = bytearray()
    for name, value in headers:
        out += _hpack_int(0, prefix, first)  # literal with name literal
        n = name.encode()
        v = value.encode()
        out += _hpack_int(len(n), 7, 0x00) + n  # not huffman
        out += _hpack_int(len(v), 7, 0x00) + v
    return bytes(out)


def hpack_table_size_update(size: int) -> bytes:
    return _hpack_int(size, 5, 0x20)


def grpc_frame(msg: bytes, compressed: int = 0) -> bytes:
    return bytes([compressed]) + struct.pack("!I", len(msg)) + msg


class StreamState:
    def __init__(self):
        self.headers: List[Tuple[str, str]] = []
        self.trailers: List[Tuple[str, str]] = []
        self.data = b""
        self.header_blocks = 0
        self.ended = False
        self.rst: Optional[int] = None

    def grpc_status(self) -> Optional[str]:
        for n, v in self.trailers + self.headers:
            if n == "grpc-status":
                return v
        return None

    def grpc_messages(self) -> List[bytes]:
        out = []
        buf = self.data
        while len(buf) >= 5:
            ln = struct.unpack("!I", buf[1:5])[0]
            if len(buf) < 5 + ln:
                break
            out.append(buf[5:5 + ln])
            buf = buf[5 + ln:]
        return out


class H2Conn:
    """One raw HTTP/2 connection over a unix socket."""

    def __init__(self, path: str, timeout: float = 10.0,
                 settings: Optional[List[Tuple[int, int]]] = None,
                 handshake: bool = True):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(timeout)
        self.sock.connect(path)
        self.streams: Dict[int, StreamState] = {}
        self.pings_acked: List[bytes] = []
        self.server_settings: List[Tuple[int, int]] = []
        self.goaway: Optional[Tuple[int, int, bytes]] = None
        self.settings_acked = False
        self._next_stream = 1
        self._buf = b""
        self._hdr_frag: Dict[int, bytes] = {}
        self._hdr_flags: Dict[int, int] = {}
        from ..native import _load

        self._dec = _load("_h2tool").HpackDecoder()
        self._enc = None
        if handshake:
            payload = b"".join(
                struct.pack("!HI", i, v) for i, v in (settings or [])
            )
            self.sock.sendall(PREFACE + frame(SETTINGS, 0, 0, payload))

    # ---- send side ----

    def send_raw(self, data: bytes) -> None:
        self.sock.sendall(data)

    def next_stream_id(self) -> int:
        sid = self._next_stream
        self._next_stream += 2
        return sid

    def encode_headers(self, headers: List[Tuple[str, str]],
                       mode: str = "index") -> bytes:
        if mode == "plain":
            return hpack_plain(headers)
        if mode == "never":
            return hpack_plain(headers, never=True)
        if mode == "index":
            if self._enc is None:
                from ..native import _load

                self._enc = _load("_h2tool").HpackEncoder()
            return self._enc.encode(headers)
        raise ValueError(mode)

    def send_headers(self, stream_id: int, headers: List[Tuple[str, str]],
                     end_stream: bool = False, mode: str = "index",
                     continuation_chunks: int = 1,
                     table_update: Optional[int] = None) -> None:
        block = self.encode_headers(headers, mode)
        if table_update is not None:
            block = hpack_table_size_update(table_update) + block
        flags = FLAG_END_STREAM if end_stream else 0
        if continuation_chunks <= 1:
            self.send_raw(frame(HEADERS, flags | FLAG_END_HEADERS,
                                stream_id, block))
            return
        # split the block across HEADERS + CONTINUATION frames
        n = max(1, len(block) // continuation_chunks)
        chunks = [block[i:i + n] for i in range(0, len(block), n)]
        self.send_raw(frame(HEADERS, flags, stream_id, chunks[0]))
        for c in chunks[1:-1]:
            self.send_raw(frame(CONTINUATION, 0, stream_id, c))
        self.send_raw(frame(CONTINUATION, FLAG_END_HEADERS, stream_id,
                            chunks[-1]))

    def send_data(self, stream_id: int, payload: bytes,
                  end_stream: bool = False, pad: int = 0,
                  chunk: int = 0) -> None:
        flags = FLAG_END_STREAM if end_stream else 0
        if chunk and len(payload) > chunk:
            for i in range(0, len(payload), chunk):
                last = i + chunk >= len(payload)
                self.send_raw(frame(DATA, flags if last else 0, stream_id,
                                    payload[i:i + chunk]))
            return
        if pad:
            self.send_raw(frame(DATA, flags | FLAG_PADDED, stream_id,
                                bytes([pad]) + payload + b"\x00" * pad))
        else:
            self.send_raw(frame(DATA, flags, stream_id, payload))

    def ping(self, payload: bytes = b"\x00" * 8, ack: bool = False) -> None:
        assert len(payload) == 8
        self.send_raw(frame(PING, FLAG_ACK if ack else 0, 0, payload))

    def rst_stream(self, stream_id: int, code: int = 8) -> None:  # CANCEL
        self.send_raw(frame(RST_STREAM, 0, stream_id,
                            struct.pack("!I", code)))

    def window_update(self, increment: int, stream_id: int = 0) -> None:
        self.send_raw(frame(WINDOW_UPDATE, 0, stream_id,
                            struct.pack("!I", increment)))

    def send_goaway(self, last_stream: int = 0, code: int = 0,
                    debug: bytes = b"") -> None:
        self.send_raw(frame(GOAWAY, 0, 0,
                            struct.pack("!II", last_stream, code) + debug))

    def settings_ack(self) -> None:
        self.send_raw(frame(SETTINGS, FLAG_ACK, 0, b""))

    # ---- receive side ----

    def stream(self, sid: int) -> StreamState:
        return self.streams.setdefault(sid, StreamState())

    def _on_frame(self, ftype: int, flags: int, sid: int,
                  payload: bytes) -> None:
        if ftype == SETTINGS:
            if flags & FLAG_ACK:
                self.settings_acked = True
            else:
                self.server_settings = [
                    struct.unpack("!HI", payload[i:i + 6])
                    for i in range(0, len(payload) - 5, 6)
                ]
                self.settings_ack()
        elif ftype == PING:
            if flags & FLAG_ACK:
                self.pings_acked.append(payload)
            else:
                self.ping(payload, ack=True)
        elif ftype in (HEADERS, CONTINUATION):
            frag = payload
            if ftype == HEADERS:
                if flags & FLAG_PADDED:
                    pad = frag[0]
                    frag = frag[1:len(frag) - pad]
                self._hdr_flags[sid] = flags
                self._hdr_frag[sid] = frag
            else:
                self._hdr_frag[sid] = self._hdr_frag.get(sid, b"") + frag
            if flags & FLAG_END_HEADERS:
                block = self._hdr_frag.pop(sid)
                hflags = self._hdr_flags.pop(sid, 0)
                decoded = self._dec.decode(block)
                st = self.stream(sid)
                st.header_blocks += 1
                if st.header_blocks == 1:
                    st.headers = decoded
                else:
                    st.trailers = decoded
                if hflags & FLAG_END_STREAM or (
                    ftype == HEADERS and flags & FLAG_END_STREAM
                ):
                    st.ended = True
        elif ftype == DATA:
            if flags & FLAG_PADDED:
                pad = payload[0]
                payload = payload[1:len(payload) - pad]
            st = self.stream(sid)
            st.data += payload
            if flags & FLAG_END_STREAM:
                st.ended = True
            # naive flow control: restore both windows
            if payload:
                self.window_update(len(payload), 0)
                if not st.ended:
                    self.window_update(len(payload), sid)
        elif ftype == RST_STREAM:
            st = self.stream(sid)
            st.rst = struct.unpack("!I", payload[:4])[0]
            st.ended = True
        elif ftype == GOAWAY:
            last, code = struct.unpack("!II", payload[:8])
            self.goaway = (last, code, payload[8:])

    def pump(self, timeout: float = 0.5) -> bool:
        """Read whatever arrives within `timeout`; False on clean EOF."""
        deadline = time.monotonic() + timeout
        self.sock.settimeout(max(0.05, timeout))
        alive = True
        while time.monotonic() < deadline:
            try:
                chunk = self.sock.recv(65536)
            except socket.timeout:
                break
            except OSError:
                alive = False
                break
            if not chunk:
                alive = False
                break
            self._buf += chunk
            while len(self._buf) >= 9:
                ln = struct.unpack("!I", b"\x00" + self._buf[:3])[0]
                if len(self._buf) < 9 + ln:
                    break
                ftype, flags = self._buf[3], self._buf[4]
                sid = struct.unpack("!I", self._buf[5:9])[0] & 0x7FFFFFFF
                payload = self._buf[9:9 + ln]
                self._buf = self._buf[9 + ln:]
                self._on_frame(ftype, flags, sid, payload)
            # return early once quiet? keep simple: read until timeout
            self.sock.settimeout(
                max(0.05, deadline - time.monotonic())
            )
        return alive

    def wait(self, cond, timeout: float = 5.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if cond():
                return True
            if not self.pump(0.2):
                return cond()
        return cond()

    def close(self) -> None:
        try:
            self.sock.close()
        except OSError:
            pass

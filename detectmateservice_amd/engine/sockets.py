"""Framed pair-socket transport (the NNG replacement at pipeline edges).

The reference uses pynng Pair0 for every hop (engine_socket.py:38-78,
engine.py:148). This framework's intra-node fast path is RCCL over xGMI
(``detectmateservice_amd.parallel``); the edge transport here carries the
same capability surface as the reference's socket layer without NNG:

* schemes ``ipc://`` (unix domain socket; stale file unlinked before bind —
  reference engine_socket.py:46-54), ``tcp://host:port``,
  ``tls+tcp://host:port`` (stdlib ``ssl``; server cert+key from one PEM as
  the reference's ``cert_key_file``, client CA + SNI as ``ca_file`` /
  ``server_name`` — engine_socket.py:60-73, engine.py:156-170),
  ``ws://`` (real RFC 6455 WebSocket framing), ``inproc://``
  (process-local queue pair), and ``shm://`` (shared-memory ring under
  /dev/shm for co-located services — zero kernel copies, in-place
  packed decode; ops/csrc/shm_ring.cpp).
* a listener (stage input) that accepts peers and can reply to the sender
  of the last received frame (the reference's request/reply fallback mode,
  engine.py:248-264),
* dialers (stage outputs) that connect non-blocking, reconnect in the
  background forever (engine.py:173-175), buffer a bounded number of frames
  (``engine_buffer_size``, engine.py:153-154) and report send failure so the
  engine can retry-then-drop (engine.py:281-301).

Wire framing: ``tcp://`` and ``tls+tcp://`` speak the NNG SP mapping
(8-octet ``\x00SP\x00`` + pair0 protocol-id connection header, then
64-bit BE length-prefixed messages) so NNG/pynng/fluentd-nng peers
interoperate at the pipeline edges; ``ws://`` carries real RFC 6455
binary frames (upgrade handshake, masked client frames, ping/pong);
``ipc://`` and ``inproc://`` are intra-node and use a compact 4-byte
length prefix. These framings plus the proto3 schemas in
``detectmateservice_amd.schemas`` are the complete wire contract between
stages.
"""
from __future__ import annotations

import logging
import os
import queue
import socket
import ssl
import struct
import threading
import time
from pathlib import Path
from typing import Dict, List, Optional, Protocol, Tuple, runtime_checkable

from ..settings import EngineAddr, TlsInputConfig, TlsOutputConfig

MAX_FRAME_BYTES = 64 * 1024 * 1024
_LEN = struct.Struct(">I")

#: NNG/nanomsg SP wire compatibility for tcp:// and tls+tcp:// (the
#: reference's fluentd plugins dial NNG pair0 over TCP — SURVEY.md §2.4):
#: connections open with the 8-octet SP header ``\x00SP\x00`` + 16-bit BE
#: protocol id + 2 reserved octets, and every message is prefixed with a
#: 64-bit BE length (SP TCP mapping). ipc:// and inproc:// are intra-node
#: and use this framework's compact 4-byte framing.
SP_PAIR0_PROTO = 0x10
_LEN64 = struct.Struct(">Q")


def _tune_buffers(sock: socket.socket) -> None:
    """Best-effort 4 MB kernel buffers: batched senders write ~1 MB blobs
    per sendall; the 208 KB default forces sender/reader lockstep."""
    for opt in (socket.SO_SNDBUF, socket.SO_RCVBUF):
        try:
            sock.setsockopt(socket.SOL_SOCKET, opt, 4 << 20)
        except OSError:
            pass


def _sp_header_bytes(proto: int = SP_PAIR0_PROTO) -> bytes:
    return b"\x00SP\x00" + struct.pack(">H", proto) + b"\x00\x00"


class RecvTimeout(Exception):
    """recv() deadline expired with no frame available."""


class SocketClosed(Exception):
    """Operation on a closed socket."""


@runtime_checkable
class EngineSocket(Protocol):
    """Transport contract the engine depends on (reference engine_socket.py:12-20)."""

    def recv(self, timeout_ms: Optional[int] = None) -> bytes: ...
    def send(self, data: bytes, block: bool = True) -> bool: ...
    def close(self) -> None: ...


# ---------------------------------------------------------------------------
# framing helpers
# ---------------------------------------------------------------------------


def _send_frame(sock: socket.socket, data: bytes, sp: bool = False) -> None:
    if len(data) > MAX_FRAME_BYTES:
        raise ValueError(f"frame of {len(data)} bytes exceeds MAX_FRAME_BYTES")
    if sp:
        sock.sendall(_LEN64.pack(len(data)) + data)
    else:
        sock.sendall(_LEN.pack(len(data)) + data)


def _recv_exact(sock: socket.socket, n: int) -> Optional[bytes]:
    buf = bytearray()
    while len(buf) < n:
        try:
            chunk = sock.recv(n - len(buf))
        except (ssl.SSLWantReadError, BlockingIOError):
            continue
        if not chunk:
            return None
        buf += chunk
    return bytes(buf)


def _recv_frame(sock: socket.socket, sp: bool = False) -> Optional[bytes]:
    if sp:
        header = _recv_exact(sock, 8)
        if header is None:
            return None
        (length,) = _LEN64.unpack(header)
    else:
        header = _recv_exact(sock, 4)
        if header is None:
            return None
        (length,) = _LEN.unpack(header)
    if length > MAX_FRAME_BYTES:
        raise ValueError(f"peer announced oversize frame ({length} bytes)")
    if length == 0:
        return b""
    return _recv_exact(sock, length)


class _FrameReader:
    """Buffered frame parser: large recv() chunks sliced into frames (a
    per-frame header+payload recv pair measured ~13k frames/s; chunked
    reads remove the per-frame syscall cost)."""

    __slots__ = ("sock", "sp", "buf")

    def __init__(self, sock: socket.socket, sp: bool) -> None:
        self.sock = sock
        self.sp = sp
        self.buf = bytearray()

    def next_frames(self) -> Optional[List[bytes]]:
        """Block for at least one frame, return ALL complete frames
        buffered so far (one queue hand-off per recv chunk, not per
        frame). None = peer closed."""
        hdr = 8 if self.sp else 4
        while True:
            frames: List[bytes] = []
            pos = 0
            buf = self.buf
            while len(buf) - pos >= hdr:
                length = int.from_bytes(buf[pos:pos + hdr], "big")
                if length > MAX_FRAME_BYTES:
                    raise ValueError(f"peer announced oversize frame ({length})")
                if len(buf) - pos < hdr + length:
                    break
                frames.append(bytes(buf[pos + hdr:pos + hdr + length]))
                pos += hdr + length
            if pos:
                del buf[:pos]
            if frames:
                return frames
            try:
                chunk = self.sock.recv(262144)
            except (ssl.SSLWantReadError, BlockingIOError):
                continue
            if not chunk:
                return None
            buf += chunk

    def next_frame(self) -> Optional[bytes]:
        frames = self.next_frames()
        if frames is None:
            return None
        # only used by single-frame consumers; re-buffer the rest
        if len(frames) > 1:
            hdr = 8 if self.sp else 4
            rest = bytearray()
            for f in frames[1:]:
                rest += len(f).to_bytes(hdr, "big") + f
            self.buf = rest + self.buf
        return frames[0]


# ---------------------------------------------------------------------------
# ws:// — minimal RFC 6455 framing (binary messages, ping/pong, close).
# The reference supports ws as an NNG transport scheme (settings.py:30-37
# there); here ws carries the same pair semantics over real WebSocket
# frames so browser/proxy-facing edges interoperate.
# ---------------------------------------------------------------------------

_WS_GUID = b"258EAFA5-E914-47DA-95CA-C5AB0DC85B11"


def _ws_accept_key(key: bytes) -> bytes:
    import base64
    import hashlib

    return base64.b64encode(hashlib.sha1(key + _WS_GUID).digest())


def _ws_server_handshake(sock: socket.socket) -> bool:
    """Read the HTTP upgrade request, reply 101. Blocking, pre-reader."""
    try:
        buf = b""
        while b"\r\n\r\n" not in buf:
            chunk = sock.recv(4096)
            if not chunk:
                return False
            buf += chunk
            if len(buf) > 16384:
                return False
        headers = {}
        for line in buf.split(b"\r\n")[1:]:
            if b":" in line:
                k, v = line.split(b":", 1)
                headers[k.strip().lower()] = v.strip()
        key = headers.get(b"sec-websocket-key")
        if key is None:
            return False
        resp = (
            b"HTTP/1.1 101 Switching Protocols\r\n"
            b"Upgrade: websocket\r\nConnection: Upgrade\r\n"
            b"Sec-WebSocket-Accept: " + _ws_accept_key(key) + b"\r\n\r\n"
        )
        sock.sendall(resp)
        return True
    except OSError:
        return False


def _ws_client_handshake(sock: socket.socket, host: str) -> bool:
    import base64
    import os as _os

    try:
        key = base64.b64encode(_os.urandom(16))
        req = (
            b"GET / HTTP/1.1\r\nHost: " + host.encode() + b"\r\n"
            b"Upgrade: websocket\r\nConnection: Upgrade\r\n"
            b"Sec-WebSocket-Key: " + key + b"\r\nSec-WebSocket-Version: 13\r\n\r\n"
        )
        sock.sendall(req)
        buf = b""
        while b"\r\n\r\n" not in buf:
            chunk = sock.recv(4096)
            if not chunk:
                return False
            buf += chunk
            if len(buf) > 16384:
                return False
        if b" 101 " not in buf.split(b"\r\n", 1)[0]:
            return False
        expect = _ws_accept_key(key)
        return expect in buf
    except OSError:
        return False


def _ws_control(opcode: int, payload: bytes, mask: bool) -> bytes:
    import os as _os

    header = bytearray([0x80 | opcode])
    mask_bit = 0x80 if mask else 0
    header.append(mask_bit | len(payload))
    if mask:
        mkey = _os.urandom(4)
        header += mkey
        payload = bytes(b ^ mkey[i % 4] for i, b in enumerate(payload))
    return bytes(header) + payload


def _ws_xor_mask(data: bytes, mkey: bytes) -> bytes:
    """Vectorized WebSocket masking (the per-byte Python loop measured
    ~20k lines/s end to end — tools/bench_transport.py ws): big-int XOR
    for typical log-frame sizes (no per-call numpy overhead), numpy for
    large payloads."""
    n = len(data)
    if n == 0:
        return data
    if n <= 4096:
        key = mkey * ((n >> 2) + 1)
        return (int.from_bytes(data, "little")
                ^ int.from_bytes(key[:n], "little")).to_bytes(n, "little")
    import numpy as _np

    arr = _np.frombuffer(data, dtype=_np.uint8)
    key = _np.frombuffer((mkey * ((n + 3) // 4))[:n], dtype=_np.uint8)
    return (arr ^ key).tobytes()


def _ws_encode(data: bytes, mask: bool) -> bytes:
    """One binary message frame (FIN=1, opcode 2); clients mask."""
    import os as _os

    header = bytearray([0x82])  # FIN + binary
    n = len(data)
    mask_bit = 0x80 if mask else 0
    if n < 126:
        header.append(mask_bit | n)
    elif n < (1 << 16):
        header.append(mask_bit | 126)
        header += n.to_bytes(2, "big")
    else:
        header.append(mask_bit | 127)
        header += n.to_bytes(8, "big")
    if mask:
        mkey = _os.urandom(4)
        header += mkey
        return bytes(header) + _ws_xor_mask(data, mkey)
    return bytes(header) + data


class _WsFrameReader:
    """Parse WebSocket frames into messages; answers pings, honors close."""

    def __init__(self, sock: socket.socket, server_side: bool) -> None:
        self.sock = sock
        self.server_side = server_side
        self.buf = bytearray()
        self._off = 0  # consumed-bytes cursor (del buf[:n] per frame is O(buffer))
        self._frag: Optional[bytearray] = None  # in-progress fragmented msg

    def _avail(self) -> int:
        return len(self.buf) - self._off

    def _fill(self, need: int) -> bool:
        while self._avail() < need:
            # compact lazily: only when the consumed prefix dominates
            if self._off > 1 << 20 and self._off > len(self.buf) // 2:
                del self.buf[: self._off]
                self._off = 0
            try:
                chunk = self.sock.recv(262144)
            except (ssl.SSLWantReadError, BlockingIOError):
                continue
            if not chunk:
                return False
            self.buf += chunk
        return True

    def next_frames(self) -> Optional[List[bytes]]:
        out: List[bytes] = []
        while True:
            if not self._fill(2):
                return out or None
            o = self._off
            b0, b1 = self.buf[o], self.buf[o + 1]
            opcode = b0 & 0x0F
            masked = bool(b1 & 0x80)
            n = b1 & 0x7F
            pos = 2
            if n == 126:
                if not self._fill(4):
                    return out or None
                n = int.from_bytes(self.buf[o + 2:o + 4], "big")
                pos = 4
            elif n == 127:
                if not self._fill(10):
                    return out or None
                n = int.from_bytes(self.buf[o + 2:o + 10], "big")
                pos = 10
            if n > MAX_FRAME_BYTES:
                raise ValueError("oversize ws frame")
            need = pos + (4 if masked else 0) + n
            if not self._fill(need):
                return out or None
            o = self._off
            if masked:
                mkey = bytes(self.buf[o + pos:o + pos + 4])
                payload = _ws_xor_mask(
                    bytes(self.buf[o + pos + 4:o + need]), mkey)
            else:
                payload = bytes(self.buf[o + pos:o + need])
            self._off += need
            if opcode == 0x8:  # close
                return out or None
            if opcode == 0x9:  # ping -> pong (server unmasked, client masked)
                try:
                    self.sock.sendall(_ws_control(0xA, payload, not self.server_side))
                except OSError:
                    return out or None
                continue
            if opcode in (0x2, 0x1, 0x0):  # binary/text incl. fragmentation
                fin = bool(b0 & 0x80)
                if opcode == 0x0:  # continuation
                    if self._frag is None:
                        raise ValueError("ws continuation without start frame")
                    self._frag += payload
                    if not fin:
                        continue
                    payload, self._frag = bytes(self._frag), None
                elif not fin:  # fragmented message start
                    if self._frag is not None:
                        raise ValueError("ws nested fragmented message")
                    self._frag = bytearray(payload)
                    continue
                out.append(payload)
                if not self._avail():
                    if self._off:
                        del self.buf[:]
                        self._off = 0
                    return out
                continue
            # pong / unknown control: ignore and keep parsing
            continue


def _sp_handshake(sock: socket.socket, logger: logging.Logger) -> bool:
    """Exchange the 8-octet SP connection header (both sides send theirs;
    any peer protocol id is accepted — pair0 expected)."""
    try:
        sock.sendall(_sp_header_bytes())
        peer = _recv_exact(sock, 8)
        if peer is None:
            return False
        if peer[:4] != b"\x00SP\x00":
            logger.warning("peer sent a non-SP header %r; closing", peer)
            return False
        proto = int.from_bytes(peer[4:6], "big")
        if proto != SP_PAIR0_PROTO:
            logger.debug("peer SP protocol id %d (expected pair0=%d)",
                         proto, SP_PAIR0_PROTO)
        return True
    except (OSError, ssl.SSLError):
        return False


# ---------------------------------------------------------------------------
# inproc registry
# ---------------------------------------------------------------------------

_INPROC_LOCK = threading.Lock()
_INPROC_ENDPOINTS: Dict[str, "InprocListener"] = {}


# ---------------------------------------------------------------------------
# Listener (stage input)
# ---------------------------------------------------------------------------


class PairListener:
    """Bound input socket: accepts peers, queues inbound frames.

    ``recv`` pops from a shared queue fed by one reader thread per peer;
    ``send`` replies to the peer whose frame was received last (request/
    reply compatibility mode, reference engine.py:248-264).
    """

    def __init__(
        self,
        addr: str,
        logger: Optional[logging.Logger] = None,
        tls_config: Optional[TlsInputConfig] = None,
        buffer_size: int = 128,
    ) -> None:
        self.addr = EngineAddr.validate(addr)
        self._log = logger or logging.getLogger(__name__)
        self._recv_q: "queue.Queue[Tuple[socket.socket, bytes]]" = queue.Queue(
            maxsize=max(buffer_size, 1) * 4
        )
        self._peers: List[socket.socket] = []
        self._peers_lock = threading.Lock()
        self._last_sender: Optional[socket.socket] = None
        self._closed = threading.Event()
        self._ssl_ctx: Optional[ssl.SSLContext] = None
        #: packed fast path (enable_packed): reader threads decode
        #: LogSchema frames straight into tensors — no Python objects
        self._packed: Optional[Tuple[int, bool, int]] = None
        self._reader_threads: List[threading.Thread] = []
        # Resolve the native reader ONCE here (synchronously): a lazy
        # import inside the reader thread costs a full torch import on a
        # cold process (~2 s) and delays the first frames past recv
        # deadlines.
        try:
            from ..ops import _dmx_C as _native  # type: ignore[attr-defined]

            self._native_mod = _native
        except Exception:  # noqa: BLE001 - extension absent: python readers
            self._native_mod = None

        scheme, rest = self.addr.scheme, self.addr.rest
        #: SP (NNG pair0) wire mapping on tcp/tls+tcp edges
        self._sp = scheme in ("tcp", "tls+tcp")
        #: real RFC 6455 framing on ws:// edges
        self._ws = scheme == "ws"
        if scheme == "inproc":
            raise ValueError("use InprocListener for inproc:// addresses")
        if scheme == "ipc":
            path = Path(rest if rest.startswith("/") else "/" + rest)
            # Unlink a stale socket file before binding (reference
            # engine_socket.py:46-54).
            try:
                if path.exists():
                    path.unlink()
            except OSError as exc:
                self._log.warning("could not unlink stale ipc socket %s: %s", path, exc)
            self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            self._sock.bind(str(path))
        else:  # tcp / tls+tcp / ws
            host, port = rest.rsplit(":", 1)
            host = host.strip("[]") or "0.0.0.0"
            self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._sock.bind((host, int(port)))
            if scheme == "tls+tcp":
                if tls_config is None:
                    raise ValueError("tls+tcp listener requires tls_input config")
                # TLS context configured BEFORE listen (reference invariant,
                # test_tls_transport.py:156-189).
                ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
                ctx.load_cert_chain(certfile=str(tls_config.cert_key_file))
                self._ssl_ctx = ctx
        self._sock.listen(8)
        self._sock.settimeout(0.2)
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name=f"PairListenerAccept[{addr}]", daemon=True
        )
        self._accept_thread.start()

    @property
    def bound_port(self) -> Optional[int]:
        """Actual TCP port (useful when bound to port 0 in tests)."""
        try:
            return self._sock.getsockname()[1]
        except (OSError, IndexError, TypeError):
            return None

    def _accept_loop(self) -> None:
        while not self._closed.is_set():
            try:
                conn, _ = self._sock.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            _tune_buffers(conn)
            if self._ssl_ctx is not None:
                try:
                    conn = self._ssl_ctx.wrap_socket(conn, server_side=True)
                except ssl.SSLError as exc:
                    self._log.warning("TLS handshake failed: %s", exc)
                    conn.close()
                    continue
            if self._sp and not _sp_handshake(conn, self._log):
                conn.close()
                continue
            if self._ws and not _ws_server_handshake(conn):
                conn.close()
                continue
            with self._peers_lock:
                self._peers.append(conn)
            rt = threading.Thread(
                target=self._reader_loop, args=(conn,),
                name="PairListenerReader", daemon=True,
            )
            self._reader_threads.append(rt)
            rt.start()

    def _reader_loop(self, conn: socket.socket) -> None:
        # Native C++ receive loop for plain fd sockets (GIL released around
        # poll/recv/parse — frame_reader.cpp); Python readers for TLS/ws.
        native = None
        if not self._ws and self._ssl_ctx is None and self._native_mod is not None:
            try:
                native = self._native_mod.FdFrameReader(conn.fileno(), self._sp)
            except Exception:  # noqa: BLE001 - bad fd etc: python path
                native = None
        try:
            if native is not None:
                while not self._closed.is_set():
                    packed = self._packed
                    if packed is not None:
                        max_len, pin, max_frames = packed
                        chunk = native.read_batch_packed(
                            max_frames, 200, max_len, pin)
                        if chunk[0].shape[0] > 0:
                            # bounded put: never deadlock a close() join
                            while not self._closed.is_set():
                                try:
                                    self._recv_q.put((conn, chunk),
                                                     timeout=0.5)
                                    break
                                except queue.Full:
                                    continue
                        continue
                    frames = native.read_batch(4096, 200)
                    if frames:
                        self._recv_q.put((conn, frames))
            else:
                reader = (_WsFrameReader(conn, server_side=True) if self._ws
                          else _FrameReader(conn, self._sp))
                while not self._closed.is_set():
                    frames = reader.next_frames()
                    if frames is None:
                        break
                    self._recv_q.put((conn, frames))
        except (OSError, ValueError, EOFError, RuntimeError):
            pass
        finally:
            with self._peers_lock:
                if conn in self._peers:
                    self._peers.remove(conn)
            try:
                conn.close()
            except OSError:
                pass

    def _pop_pending(self) -> Optional[bytes]:
        if getattr(self, "_pending", None):
            return self._pending.pop(0)
        return None

    def recv(self, timeout_ms: Optional[int] = None) -> bytes:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        frame = self._pop_pending()
        if frame is not None:
            return frame
        try:
            if timeout_ms is None:
                conn, frames = self._recv_q.get()
            else:
                conn, frames = self._recv_q.get(timeout=timeout_ms / 1000.0)
        except queue.Empty:
            raise RecvTimeout(self.addr) from None
        self._last_sender = conn
        # keep _pending_conns aligned with _pending: recv() and recv_many()
        # may be interleaved, and reply(idx) routes by _batch_conns built
        # from these lists — a bare _pending refill would desync them
        self._pending = list(frames[1:])
        self._pending_conns = [conn] * len(self._pending)
        return frames[0]

    def has_pending(self) -> bool:
        return bool(getattr(self, "_pending", None)) or not self._recv_q.empty()

    # -- packed fast path ----------------------------------------------
    def enable_packed(self, max_len: int, pin: bool,
                      max_frames: int = 4096) -> bool:
        """Switch reader threads to the native socket→tensor decode path
        (plain fd sockets only; TLS/ws readers are unaffected and keep
        delivering byte frames)."""
        if self._ws or self._ssl_ctx is not None or self._native_mod is None:
            return False
        self._packed = (max_len, pin, max_frames)
        # Each queued chunk can hold a full [max_frames, max_len] (pinned)
        # buffer (~2 MB at 8192x256): bound the queue so a fast producer
        # backpressures through the socket instead of ballooning pinned
        # host memory.
        self._recv_q.maxsize = 64
        return True

    def _as_packed(self, item):
        """Normalize a queue item to (lines, lens, blob, off, nbytes) —
        byte-frame items (queued before enable_packed, or from TLS/ws
        peers) are converted via the batch codec."""
        import torch

        if len(item) == 5 and torch.is_tensor(item[0]):
            lines, lens, blob, off, nbytes = item
            return lines, lens, blob, off, int(nbytes)
        max_len, pin = (self._packed or (256, False, 0))[:2]
        frames = [f for f in item if f]
        nbytes = sum(len(f) for f in frames)
        lines, lens, blob, off = self._native_mod.decode_log_batch_packed(
            list(frames), max_len, pin
        )
        return lines, lens, blob, off, nbytes

    def recv_packed(self, timeout_ms: int, max_frames: int = 0,
                    linger_ms: float = 0.0):
        """Pop packed chunks, merging up to ``max_frames`` rows within
        ``linger_ms``: (conn, lines, lens, ids_blob, ids_off, frame_bytes).
        ``conn`` is the FIRST chunk's sender; per-row routing for merged
        multi-peer batches is available via ``reply_row`` — each chunk's
        (conn, row_count) is recorded in ``_batch_conn_segs``."""
        import torch

        try:
            conn, item = self._recv_q.get(timeout=timeout_ms / 1000.0)
        except queue.Empty:
            raise RecvTimeout(self.addr) from None
        self._last_sender = conn
        lines, lens, blob, off, nbytes = self._as_packed(item)
        if max_frames <= 0 or lines.shape[0] >= max_frames:
            self._batch_conn_segs = [(conn, int(lines.shape[0]))]
            return conn, lines, lens, blob, off, nbytes
        chunks = [(lines, lens, blob, off, nbytes)]
        segs = [(conn, int(lines.shape[0]))]
        total = int(lines.shape[0])
        deadline = time.monotonic() + linger_ms / 1000.0
        while total < max_frames:
            remaining = deadline - time.monotonic()
            try:
                c, item = self._recv_q.get(
                    timeout=max(remaining, 0) if remaining > 0 else None,
                    block=remaining > 0,
                )
            except queue.Empty:
                break
            ch = self._as_packed(item)
            chunks.append(ch)
            segs.append((c, int(ch[0].shape[0])))
            total += int(ch[0].shape[0])
        self._batch_conn_segs = segs
        if len(chunks) == 1:
            return conn, lines, lens, blob, off, nbytes
        lines = torch.cat([c[0] for c in chunks])
        lens = torch.cat([c[1] for c in chunks])
        blob = b"".join(c[2] for c in chunks)
        offs = [chunks[0][3]]
        base = int(chunks[0][3][-1])
        for c in chunks[1:]:
            offs.append(c[3][1:] + base)
            base += int(c[3][-1])
        off = torch.cat(offs)
        nbytes = sum(c[4] for c in chunks)
        return conn, lines, lens, blob, off, nbytes

    def recv_many(
        self, max_frames: int, timeout_ms: int, linger_ms: float = 0.0
    ) -> List[bytes]:
        """Drain up to ``max_frames`` frames: wait up to ``timeout_ms`` for the
        first, then keep draining without waiting more than ``linger_ms`` total.

        This is the batched-engine entry point (SURVEY.md §7 design
        departures) — no reference equivalent. ``_batch_conns`` records
        each frame's source connection so the request/reply mode routes
        every reply to ITS sender even with several concurrent peers
        (the reference's Pair0 was strictly 1:1 and never faced this).
        """
        out: List[bytes] = list(getattr(self, "_pending", None) or [])
        conns: List = list(getattr(self, "_pending_conns", None) or [])
        self._pending = []
        self._pending_conns = []
        if len(out) >= max_frames:
            self._pending = out[max_frames:]
            self._pending_conns = conns[max_frames:]
            self._batch_conns = conns[:max_frames]
            return out[:max_frames]
        try:
            if not out:
                conn, frames = self._recv_q.get(timeout=timeout_ms / 1000.0)
                self._last_sender = conn
                out.extend(frames)
                conns.extend([conn] * len(frames))
        except queue.Empty:
            self._batch_conns = conns
            return out
        deadline = time.monotonic() + linger_ms / 1000.0
        while len(out) < max_frames:
            remaining = deadline - time.monotonic()
            try:
                conn, frames = self._recv_q.get(
                    timeout=max(remaining, 0) if remaining > 0 else None,
                    block=remaining > 0,
                )
            except queue.Empty:
                break
            self._last_sender = conn
            out.extend(frames)
            conns.extend([conn] * len(frames))
        if len(out) > max_frames:
            self._pending = out[max_frames:]
            self._pending_conns = conns[max_frames:]
            out = out[:max_frames]
            conns = conns[:max_frames]
        self._batch_conns = conns
        return out

    def reply(self, idx: int, data: bytes) -> bool:
        """Reply to the sender of frame ``idx`` of the last recv_many batch
        (falls back to the last sender when unknown)."""
        conns = getattr(self, "_batch_conns", None)
        conn = conns[idx] if conns and 0 <= idx < len(conns) else self._last_sender
        if conn is None:
            return False
        try:
            if self._ws:
                conn.sendall(_ws_encode(data, mask=False))
            else:
                _send_frame(conn, data, self._sp)
            return True
        except OSError:
            return False

    def reply_row(self, idx: int, data: bytes) -> bool:
        """Reply to the sender of ROW ``idx`` of the last recv_packed batch
        (merged multi-peer chunks route per-row via the recorded
        (conn, row_count) segments; falls back to the last sender)."""
        conn = None
        segs = getattr(self, "_batch_conn_segs", None)
        if segs:
            base = 0
            for c, rows in segs:
                if idx < base + rows:
                    conn = c
                    break
                base += rows
        return self.reply_conn(conn if conn is not None else self._last_sender,
                               data)

    def send(self, data: bytes, block: bool = True) -> bool:
        """Reply to the most recent sender (request/reply fallback mode)."""
        return self.reply_conn(self._last_sender, data)

    def reply_conn(self, conn, data: bytes) -> bool:
        """Reply to a SPECIFIC peer (the packed loop routes each batch's
        replies to that batch's sender; per-row routing for merged
        multi-peer chunks is ``reply_row``)."""
        if conn is None:
            with self._peers_lock:
                conn = self._peers[-1] if self._peers else None
        if conn is None:
            return False
        try:
            if self._ws:
                conn.sendall(_ws_encode(data, mask=False))
            else:
                _send_frame(conn, data, self._sp)
            return True
        except OSError:
            return False

    def close(self) -> None:
        self._closed.set()
        try:
            self._sock.close()
        except OSError:
            pass
        with self._peers_lock:
            for conn in self._peers:
                try:
                    conn.shutdown(socket.SHUT_RDWR)
                except OSError:
                    pass
                try:
                    conn.close()
                except OSError:
                    pass
            self._peers.clear()
        # join reader threads: a daemon thread still inside the C++
        # reader when the interpreter finalizes would be killed at GIL
        # reacquisition (pthread_exit unwinding C++ frames -> terminate)
        for rt in self._reader_threads:
            rt.join(timeout=1.0)
        self._reader_threads.clear()
        if self.addr.scheme == "ipc":
            try:
                os.unlink("/" + self.addr.rest.lstrip("/"))
            except OSError:
                pass


class InprocListener:
    """Process-local pair endpoint (reference's ``inproc://`` scheme)."""

    def __init__(self, addr: str, buffer_size: int = 128) -> None:
        self.addr = EngineAddr.validate(addr)
        self._recv_q: "queue.Queue[bytes]" = queue.Queue(maxsize=max(buffer_size, 1) * 4)
        self._reply_q: "queue.Queue[bytes]" = queue.Queue(maxsize=max(buffer_size, 1) * 4)
        self._closed = threading.Event()
        with _INPROC_LOCK:
            _INPROC_ENDPOINTS[str(self.addr)] = self

    def recv(self, timeout_ms: Optional[int] = None) -> bytes:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        try:
            if timeout_ms is None:
                return self._recv_q.get()
            return self._recv_q.get(timeout=timeout_ms / 1000.0)
        except queue.Empty:
            raise RecvTimeout(self.addr) from None

    def recv_many(self, max_frames: int, timeout_ms: int, linger_ms: float = 0.0) -> List[bytes]:
        frames: List[bytes] = []
        try:
            frames.append(self.recv(timeout_ms=timeout_ms))
        except RecvTimeout:
            return frames
        while len(frames) < max_frames:
            try:
                frames.append(self._recv_q.get_nowait())
            except queue.Empty:
                break
        return frames

    def send(self, data: bytes, block: bool = True) -> bool:
        try:
            self._reply_q.put(data, block=False)
            return True
        except queue.Full:
            return False

    def close(self) -> None:
        self._closed.set()
        with _INPROC_LOCK:
            if _INPROC_ENDPOINTS.get(str(self.addr)) is self:
                del _INPROC_ENDPOINTS[str(self.addr)]


# ---------------------------------------------------------------------------
# Dialer (stage output)
# ---------------------------------------------------------------------------


class PairDialer:
    """Output socket: non-blocking dial, background reconnect, bounded buffer.

    Elasticity parity with the reference: the downstream
    peer may be absent at startup or die mid-run; ``send(block=False)``
    returns False when the bounded buffer is full (peer down or slow) and
    the engine applies its retry-then-drop policy (engine.py:281-301).
    """

    REDIAL_INTERVAL_S = 0.25

    def __init__(
        self,
        addr: str,
        logger: Optional[logging.Logger] = None,
        tls_config: Optional[TlsOutputConfig] = None,
        buffer_size: int = 128,
        dial_timeout_s: float = 1.0,
    ) -> None:
        self.addr = EngineAddr.validate(addr)
        self._log = logger or logging.getLogger(__name__)
        self._dial_timeout_s = max(dial_timeout_s, 0.05)
        self._send_q: "queue.Queue[bytes]" = queue.Queue(maxsize=max(buffer_size, 1))
        self._recv_q: "queue.Queue[bytes]" = queue.Queue(maxsize=max(buffer_size, 1) * 4)
        self._closed = threading.Event()
        self._connected = threading.Event()
        self._tls_config = tls_config
        self._conn: Optional[socket.socket] = None
        self._inproc: Optional[InprocListener] = None
        self._sp = self.addr.scheme in ("tcp", "tls+tcp")
        self._ws = self.addr.scheme == "ws"
        self._inflight: Optional[bytes] = None
        self._send_stats = ([0, 0.0, 0.0, 0] if os.environ.get(
            "DMX_ENGINE_STATS") == "1" else None)
        try:
            from ..ops import _dmx_C as _native  # type: ignore[attr-defined]

            self._native_mod = _native
        except Exception:  # noqa: BLE001 - extension absent: python framing
            self._native_mod = None
        if self.addr.scheme == "inproc":
            # resolved lazily in the worker loop so late binding works
            pass
        self._worker = threading.Thread(
            target=self._run, name=f"PairDialer[{addr}]", daemon=True
        )
        self._worker.start()

    # -- connection management -----------------------------------------
    def _try_connect(self) -> bool:
        scheme, rest = self.addr.scheme, self.addr.rest
        if scheme == "inproc":
            with _INPROC_LOCK:
                self._inproc = _INPROC_ENDPOINTS.get(str(self.addr))
            return self._inproc is not None
        try:
            if scheme == "ipc":
                s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                s.settimeout(self._dial_timeout_s)
                s.connect("/" + rest.lstrip("/"))
            else:
                host, port = rest.rsplit(":", 1)
                host = host.strip("[]")
                s = socket.create_connection(
                    (host, int(port)), timeout=self._dial_timeout_s
                )
                if scheme == "tls+tcp":
                    if self._tls_config is None:
                        raise ValueError("tls+tcp dialer requires tls_output config")
                    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
                    ctx.load_verify_locations(cafile=str(self._tls_config.ca_file))
                    s = ctx.wrap_socket(
                        s, server_hostname=self._tls_config.server_name
                    )
                if self._sp and not _sp_handshake(s, self._log):
                    s.close()
                    return False
                if self._ws and not _ws_client_handshake(s, host):
                    s.close()
                    return False
            s.settimeout(None)
            _tune_buffers(s)
            self._conn = s
            threading.Thread(
                target=self._reader_loop, args=(s,),
                name="PairDialerReader", daemon=True,
            ).start()
            return True
        except (OSError, ssl.SSLError, ValueError):
            return False

    def _reader_loop(self, conn: socket.socket) -> None:
        reader = (_WsFrameReader(conn, server_side=False) if self._ws
                  else _FrameReader(conn, self._sp))
        try:
            while not self._closed.is_set():
                frames = reader.next_frames()
                if frames is None:
                    break
                for frame in frames:
                    try:
                        self._recv_q.put(frame, timeout=1.0)
                    except queue.Full:
                        pass  # drop inbound overflow on the reply channel
        except (OSError, ValueError):
            pass
        finally:
            if self._conn is conn:
                self._conn = None
                self._connected.clear()

    def _run(self) -> None:
        while not self._closed.is_set():
            if self.addr.scheme == "inproc":
                if self._inproc is None or self._inproc._closed.is_set():
                    self._inproc = None
                    if not self._try_connect():
                        time.sleep(self.REDIAL_INTERVAL_S)
                        continue
                self._connected.set()
                if self._inflight is not None:
                    batch, self._inflight = self._inflight, None
                else:
                    try:
                        batch = self._send_q.get(timeout=0.2)
                    except queue.Empty:
                        continue
                target = self._inproc
                undelivered = []
                for j, f in enumerate(batch):
                    ok = False
                    if target is not None and not target._closed.is_set():
                        try:
                            target._recv_q.put(f, timeout=1.0)
                            ok = True
                        except queue.Full:
                            pass
                    if not ok:
                        undelivered = batch[j:]
                        break
                if undelivered:
                    self._inflight = undelivered
                continue

            if self._conn is None:
                if not self._try_connect():
                    self._connected.clear()
                    time.sleep(self.REDIAL_INTERVAL_S)
                    continue
                self._connected.set()
                self._log.debug("dialer connected to %s", self.addr)
            # the in-flight slot preserves ordering across reconnects (a
            # plain re-put would append BEHIND frames queued meanwhile)
            if self._inflight is not None:
                batch, self._inflight = self._inflight, None
            else:
                try:
                    batch = self._send_q.get(timeout=0.2)
                except queue.Empty:
                    continue
            conn = self._conn
            if conn is None:
                self._inflight = batch
                continue
            # coalesce everything queued into ONE sendall (per-frame
            # sendall measured ~13k frames/s; batching removes the
            # syscall + GIL ping-pong per frame). 4096-frame cap ≈ 1 MB
            # per sendall; only backlog is drained, so idle-path latency
            # is unaffected.
            if len(batch) < 4096:
                batch = list(batch)
                while len(batch) < 4096:
                    try:
                        batch.extend(self._send_q.get_nowait())
                    except queue.Empty:
                        break
            _st = self._send_stats
            _t0 = time.perf_counter() if _st is not None else 0.0
            if self._ws:
                payload = b"".join(_ws_encode(f, mask=True) for f in batch)
            elif self._native_mod is not None:
                # C++ pack: one GIL-released pass (the per-frame Python
                # pack+join loop capped the send side at ~3.3M frames/s)
                payload = self._native_mod.pack_frames(batch, self._sp)
            else:
                hdr = _LEN64 if self._sp else _LEN
                parts = []
                for f in batch:
                    parts.append(hdr.pack(len(f)))
                    parts.append(f)
                payload = b"".join(parts)
            if _st is not None:
                _t1 = time.perf_counter()
            try:
                conn.sendall(payload)
                if _st is not None:
                    _st[0] += len(batch)
                    _st[1] += _t1 - _t0
                    _st[2] += time.perf_counter() - _t1
                    _st[3] += 1
                    if _st[3] % 64 == 0:
                        self._log.warning(
                            "[send-stats] %d frames in %d sendalls: pack "
                            "%.2fms/batch send %.2fms/batch (%.0f fr/batch)",
                            _st[0], _st[3], _st[1] * 1e3 / _st[3],
                            _st[2] * 1e3 / _st[3], _st[0] / _st[3])
            except OSError:
                self._log.debug("send to %s failed; reconnecting", self.addr)
                try:
                    conn.close()
                except OSError:
                    pass
                if self._conn is conn:
                    self._conn = None
                self._connected.clear()
                self._inflight = batch  # retried first after reconnect
        # drain on close
        conn = self._conn
        if conn is not None:
            try:
                conn.close()
            except OSError:
                pass

    # -- EngineSocket API ----------------------------------------------
    def send(self, data: bytes, block: bool = True) -> bool:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        try:
            self._send_q.put([data], block=block, timeout=5.0 if block else None)
            return True
        except queue.Full:
            return False

    def send_many(self, frames, block: bool = True, chunk: int = 256) -> int:
        """Enqueue many frames with one queue operation per `chunk`
        (per-frame queue hand-offs dominated the feeder side of service
        mode). Returns the number of frames accepted."""
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        frames = list(frames)
        accepted = 0
        for off in range(0, len(frames), chunk):
            part = frames[off:off + chunk]
            try:
                self._send_q.put(part, block=block, timeout=5.0 if block else None)
                accepted += len(part)
            except queue.Full:
                break
        return accepted

    def recv(self, timeout_ms: Optional[int] = None) -> bytes:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        src = self._inproc._reply_q if self._inproc is not None else self._recv_q
        try:
            if timeout_ms is None:
                return src.get()
            return src.get(timeout=timeout_ms / 1000.0)
        except queue.Empty:
            raise RecvTimeout(self.addr) from None

    def wait_connected(self, timeout_s: float) -> bool:
        return self._connected.wait(timeout=timeout_s)

    @property
    def pending(self) -> int:
        return self._send_q.qsize()

    def close(self) -> None:
        self._closed.set()
        self._worker.join(timeout=2.0)
        conn = self._conn
        if conn is not None:
            try:
                conn.close()
            except OSError:
                pass


# ---------------------------------------------------------------------------
# factory (reference engine_socket.py:23-32 seam, kept injectable for tests)
# ---------------------------------------------------------------------------


# ---------------------------------------------------------------------------
# shm:// — shared-memory ring transport for co-located services.
# The ipc:// socket path is bound by kernel copies (~4M lines/s at 258 B
# frames — BASELINE.md); the ring maps one /dev/shm file into both
# processes: the producer writes each frame once, the consumer's packed
# path proto-decodes IN PLACE into tensors (ops/csrc/shm_ring.cpp).
# Lifecycle: the LISTENER owns the files (unlinks on close); dialers
# attach (creating the ring if they start first, so late binding keeps
# buffered frames — reference engine.py:173-179 semantics) and re-attach
# when the listener recreated the ring.
# ---------------------------------------------------------------------------


def _shm_paths(addr: "EngineAddr"):
    rest = addr.rest
    base = rest if rest.startswith("/") else "/" + rest
    if not base.startswith("/dev/shm/"):
        base = "/dev/shm" + base
    return base + ".c2s", base + ".s2c"


def _shm_ring_bytes() -> int:
    return int(os.environ.get("DMX_SHM_RING_BYTES", str(32 << 20)))


class ShmListener:
    """Bound shm input: c2s ring for data, s2c ring for replies."""

    def __init__(self, addr: str, logger=None, buffer_size: int = 128,
                 tls_config=None) -> None:
        if tls_config is not None:
            raise ValueError("shm:// does not support TLS")
        self.addr = EngineAddr.validate(addr)
        self._log = logger or logging.getLogger(__name__)
        from ..ops import _dmx_C  # extension required for shm

        self._c2s_path, self._s2c_path = _shm_paths(self.addr)
        # Stale-ring detection: a marker file exists while a listener is
        # live and is removed by close(). Marker present at startup =>
        # the previous listener CRASHED and the ring files hold that
        # run's frames — recreate them fresh (ShmRing create=True
        # unlinks first). No marker => any existing rings were created
        # by an early dialer buffering ahead of us (late binding,
        # reference engine.py:173-179): attach and keep their frames.
        self._marker_path = self._c2s_path + ".live"
        stale = os.path.exists(self._marker_path)
        self._c2s = _dmx_C.ShmRing(self._c2s_path, _shm_ring_bytes(), stale)
        self._s2c = _dmx_C.ShmRing(self._s2c_path, _shm_ring_bytes(), stale)
        with open(self._marker_path, "w") as fh:
            fh.write(str(os.getpid()))
        self._pending: List[bytes] = []
        self._packed: Optional[Tuple[int, bool, int]] = None
        self._closed = threading.Event()

    # -- frame path ----------------------------------------------------
    def recv(self, timeout_ms: Optional[int] = None) -> bytes:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        if self._pending:
            return self._pending.pop(0)
        frames = self._c2s.read_batch(
            4096, 100 if timeout_ms is None else timeout_ms)
        while not frames and timeout_ms is None:
            if self._closed.is_set():
                raise SocketClosed(self.addr)
            frames = self._c2s.read_batch(4096, 100)
        if not frames:
            raise RecvTimeout(self.addr)
        self._pending = [bytes(f) for f in frames[1:]]
        return bytes(frames[0])

    def recv_many(self, max_frames: int, timeout_ms: int,
                  linger_ms: float = 0.0) -> List[bytes]:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        out = self._pending[:max_frames]
        self._pending = self._pending[max_frames:]
        deadline = time.monotonic() + linger_ms / 1000.0
        first = True
        while len(out) < max_frames:
            if out and time.monotonic() >= deadline:
                break
            tmo = timeout_ms if first and not out else max(
                1, int((deadline - time.monotonic()) * 1000))
            frames = self._c2s.read_batch(max_frames - len(out), tmo)
            first = False
            if not frames:
                break
            out.extend(bytes(f) for f in frames)
        return out

    # -- packed fast path ----------------------------------------------
    def enable_packed(self, max_len: int, pin: bool,
                      max_frames: int = 4096) -> bool:
        self._packed = (max_len, pin, max_frames)
        return True

    def recv_packed(self, timeout_ms: int, max_frames: int = 0,
                    linger_ms: float = 0.0):
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        max_len, pin, mf = self._packed or (256, False, 4096)
        if max_frames > 0:
            mf = max_frames
        lines, lens, blob, off, nbytes = self._c2s.read_batch_packed(
            mf, timeout_ms, max_len, pin)
        if lines.shape[0] == 0:
            raise RecvTimeout(self.addr)
        return None, lines, lens, blob, off, int(nbytes)

    def has_pending(self) -> bool:
        return bool(self._pending) or self._c2s.pending() > 0

    # -- reply path (request/reply compatibility mode) -----------------
    def send(self, data: bytes, block: bool = True) -> bool:
        return self._s2c.write_frames([data]) == 1

    def reply_conn(self, conn, data: bytes) -> bool:
        """Ring replies are single-channel (s2c); conn is advisory."""
        return self.send(data)

    def reply(self, idx: int, data: bytes) -> bool:
        return self.send(data)

    def reply_row(self, idx: int, data: bytes) -> bool:
        """Ring replies are single-channel (s2c)."""
        return self.send(data)

    def close(self) -> None:
        self._closed.set()
        for p in (self._c2s_path, self._s2c_path, self._marker_path):
            try:
                os.unlink(p)
            except OSError:
                pass


class ShmDialer:
    """Output side of a shm ring (stage output / feeder)."""

    REATTACH_AFTER_FULL = 50

    def __init__(self, addr: str, logger=None, buffer_size: int = 128,
                 tls_config=None, dial_timeout_s: float = 1.0) -> None:
        if tls_config is not None:
            raise ValueError("shm:// does not support TLS")
        self.addr = EngineAddr.validate(addr)
        self._log = logger or logging.getLogger(__name__)
        self._c2s_path, self._s2c_path = _shm_paths(self.addr)
        self._closed = threading.Event()
        self._full_streak = 0
        self._attach()

    def _attach(self) -> None:
        from ..ops import _dmx_C

        self._c2s = _dmx_C.ShmRing(self._c2s_path, _shm_ring_bytes(), False)
        self._s2c = _dmx_C.ShmRing(self._s2c_path, _shm_ring_bytes(), False)
        try:
            self._ino = os.stat(self._c2s_path).st_ino
        except OSError:
            self._ino = None
        self._pending_in: List[bytes] = []

    def _maybe_reattach(self) -> None:
        """The listener owns the files: if it was restarted (new inode),
        remap — mirrors the socket dialer's background reconnect."""
        self._full_streak += 1
        if self._full_streak < self.REATTACH_AFTER_FULL:
            return
        self._full_streak = 0
        try:
            ino = os.stat(self._c2s_path).st_ino
        except OSError:
            return  # listener gone entirely: keep dropping
        if ino != self._ino:
            self._log.debug("shm ring recreated; re-attaching %s", self.addr)
            self._attach()

    def wait_connected(self, timeout_s: float) -> bool:
        return True  # the ring buffers; drop-when-full supplies elasticity

    def send(self, data: bytes, block: bool = True) -> bool:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        deadline = time.monotonic() + 5.0
        while True:
            if self._c2s.write_frames([data]) == 1:
                self._full_streak = 0
                return True
            self._maybe_reattach()
            if not block or time.monotonic() >= deadline:
                return False
            time.sleep(0.0005)

    def send_many(self, frames, block: bool = True, chunk: int = 4096) -> int:
        if self._closed.is_set():
            raise SocketClosed(self.addr)
        frames = list(frames)
        accepted = 0
        while accepted < len(frames):
            n = self._c2s.write_frames(frames[accepted:accepted + chunk])
            accepted += n
            if n == 0:
                self._maybe_reattach()
                if not block:
                    break
                time.sleep(0.0005)
            else:
                self._full_streak = 0
        return accepted

    def recv(self, timeout_ms: Optional[int] = None) -> bytes:
        if self._pending_in:
            return self._pending_in.pop(0)
        frames = self._s2c.read_batch(
            4096, 100 if timeout_ms is None else timeout_ms)
        while not frames and timeout_ms is None:
            if self._closed.is_set():
                raise SocketClosed(self.addr)
            frames = self._s2c.read_batch(4096, 100)
        if not frames:
            raise RecvTimeout(self.addr)
        self._pending_in = [bytes(f) for f in frames[1:]]
        return bytes(frames[0])

    @property
    def pending(self) -> int:
        return 0

    def close(self) -> None:
        self._closed.set()


class PairSocketFactory:
    """Creates bound input sockets; injectable seam for tests/alt transports
    (reference engine.py:111-113)."""

    def create(
        self,
        addr: str,
        logger: Optional[logging.Logger] = None,
        tls_config: Optional[TlsInputConfig] = None,
        buffer_size: int = 128,
    ):
        parsed = EngineAddr.validate(addr)
        if parsed.scheme == "inproc":
            return InprocListener(addr, buffer_size=buffer_size)
        if parsed.scheme == "shm":
            return ShmListener(addr, logger=logger, buffer_size=buffer_size,
                               tls_config=tls_config)
        return PairListener(
            addr, logger=logger, tls_config=tls_config, buffer_size=buffer_size
        )

    def create_dialer(
        self,
        addr: str,
        logger: Optional[logging.Logger] = None,
        tls_config: Optional[TlsOutputConfig] = None,
        buffer_size: int = 128,
        dial_timeout_s: float = 1.0,
    ):
        parsed = EngineAddr.validate(addr)
        if parsed.scheme == "shm":
            return ShmDialer(addr, logger=logger, buffer_size=buffer_size,
                             tls_config=tls_config,
                             dial_timeout_s=dial_timeout_s)
        return PairDialer(
            addr, logger=logger, tls_config=tls_config, buffer_size=buffer_size,
            dial_timeout_s=dial_timeout_s,
        )

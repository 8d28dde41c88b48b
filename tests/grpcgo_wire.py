"""grpc-go wire-behavior simulator.

kubelet talks to the device plugin with grpc-go (golang.org/x/net/http2 +
x/net/http2/hpack), an independent HTTP/2 implementation whose on-wire
behavior differs from grpc-C-core (which tests/test_deviceplugin.py
covers) in ways that can break a hand-rolled server:

  * HPACK: huffman-codes every string when shorter, and uses INCREMENTAL
    INDEXING for almost all headers — the second RPC on a connection
    references dynamic-table entries the first inserted (grpc-C-core
    mostly emits never-indexed literals).
  * connection dance: preface + SETTINGS{MAX_FRAME_SIZE=16384}, then an
    immediate connection-level WINDOW_UPDATE; it expects its SETTINGS to
    be ACKed and ACKs the server's.
  * BDP probing: after response DATA arrives it sends a PING with the
    fixed payload {2,4,16,16,9,14,7,7} and measures the ACK round trip;
    a server that doesn't echo the exact payload stalls the transport's
    window growth.
  * keepalive PINGs (8 zero bytes) on idle connections; a missed ACK
    closes the transport.
  * flow control: returns window credit with paired stream+connection
    WINDOW_UPDATE frames after consuming DATA.
  * cancellation: RST_STREAM(CANCEL) mid-stream (e.g. kubelet restarting
    its ListAndWatch), connection stays up for later RPCs.
  * shutdown: GOAWAY(NO_ERROR) before closing the socket.

This module replays those exact behaviors against the plugin's server
(and, in reverse, plays a grpc-go *server* for the plugin's Register
client). It deliberately shares no code with the C++ stack under test —
the huffman table is read from the RFC constants header and the HPACK
static table is restated from RFC 7541 Appendix A.
"""

import re
import socket
import struct
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

# --- frame constants -------------------------------------------------------
DATA, HEADERS, PRIORITY, RST_STREAM, SETTINGS = 0x0, 0x1, 0x2, 0x3, 0x4
PING, GOAWAY, WINDOW_UPDATE, CONTINUATION = 0x6, 0x7, 0x8, 0x9
END_STREAM, ACK, END_HEADERS, PADDED, PRIORITY_FLAG = 0x1, 0x1, 0x4, 0x8, 0x20

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"
BDP_PING = bytes([2, 4, 16, 16, 9, 14, 7, 7])  # grpc-go's bdpPing payload
KEEPALIVE_PING = bytes(8)

SETTINGS_MAX_FRAME_SIZE = 0x5
SETTINGS_INITIAL_WINDOW_SIZE = 0x4


def frame(ftype, flags, stream, payload=b""):
    return (struct.pack(">I", len(payload))[1:] + bytes([ftype, flags]) +
            struct.pack(">I", stream) + payload)


def grpc_frame(msg: bytes) -> bytes:
    return b"\x00" + struct.pack(">I", len(msg)) + msg


# --- RFC 7541 Appendix B huffman table (parsed from the C++ header so the
# two sides can't drift) ----------------------------------------------------
def _load_huffman():
    text = (REPO / "native" / "grpc" / "hpack_huffman_table.h").read_text()
    pairs = re.findall(r"\{0x([0-9a-fA-F]+)u,\s*(\d+)\}", text)
    assert len(pairs) == 257, len(pairs)
    return [(int(c, 16), int(b)) for c, b in pairs]


_HUFF = _load_huffman()


def huffman_encode(data: bytes) -> bytes:
    acc, nbits = 0, 0
    out = bytearray()
    for b in data:
        code, bits = _HUFF[b]
        acc = (acc << bits) | code
        nbits += bits
        while nbits >= 8:
            nbits -= 8
            out.append((acc >> nbits) & 0xFF)
    if nbits:
        # pad with the EOS prefix (all 1s)
        out.append(((acc << (8 - nbits)) | ((1 << (8 - nbits)) - 1)) & 0xFF)
    return bytes(out)


# --- RFC 7541 Appendix A static table --------------------------------------
STATIC_TABLE = [
    (":authority", ""), (":method", "GET"), (":method", "POST"),
    (":path", "/"), (":path", "/index.html"), (":scheme", "http"),
    (":scheme", "https"), (":status", "200"), (":status", "204"),
    (":status", "206"), (":status", "304"), (":status", "400"),
    (":status", "404"), (":status", "500"), ("accept-charset", ""),
    ("accept-encoding", "gzip, deflate"), ("accept-language", ""),
    ("accept-ranges", ""), ("accept", ""), ("access-control-allow-origin", ""),
    ("age", ""), ("allow", ""), ("authorization", ""), ("cache-control", ""),
    ("content-disposition", ""), ("content-encoding", ""),
    ("content-language", ""), ("content-length", ""), ("content-location", ""),
    ("content-range", ""), ("content-type", ""), ("cookie", ""), ("date", ""),
    ("etag", ""), ("expect", ""), ("expires", ""), ("from", ""), ("host", ""),
    ("if-match", ""), ("if-modified-since", ""), ("if-none-match", ""),
    ("if-range", ""), ("if-unmodified-since", ""), ("last-modified", ""),
    ("link", ""), ("location", ""), ("max-forwards", ""),
    ("proxy-authenticate", ""), ("proxy-authorization", ""), ("range", ""),
    ("referer", ""), ("refresh", ""), ("retry-after", ""), ("server", ""),
    ("set-cookie", ""), ("strict-transport-security", ""),
    ("transfer-encoding", ""), ("user-agent", ""), ("vary", ""), ("via", ""),
    ("www-authenticate", ""),
]
assert len(STATIC_TABLE) == 61


class GoHpackEncoder:
    """Mimics x/net/http2/hpack.Encoder: exact-match -> indexed field;
    otherwise literal WITH incremental indexing (sensitive headers
    excepted), strings huffman-coded when that is shorter, entries
    inserted into a 4096-byte dynamic table with eviction."""

    def __init__(self, max_size=4096):
        self.dyn = []  # most recent first; entries (name, value)
        self.max_size = max_size

    @staticmethod
    def _entry_size(n, v):
        return len(n) + len(v) + 32

    def _evict(self):
        size = sum(self._entry_size(n, v) for n, v in self.dyn)
        while size > self.max_size and self.dyn:
            n, v = self.dyn.pop()
            size -= self._entry_size(n, v)

    def _find(self, name, value):
        """Returns (exact_index, name_index) 1-based across static+dynamic."""
        name_idx = 0
        for i, (n, v) in enumerate(STATIC_TABLE):
            if n == name:
                if v == value:
                    return i + 1, 0
                if not name_idx:
                    name_idx = i + 1
        for i, (n, v) in enumerate(self.dyn):
            if n == name:
                if v == value:
                    return 62 + i, 0
                if not name_idx:
                    name_idx = 62 + i
        return 0, name_idx

    @staticmethod
    def _int(prefix_bits, flags, value):
        limit = (1 << prefix_bits) - 1
        if value < limit:
            return bytes([flags | value])
        out = bytearray([flags | limit])
        value -= limit
        while value >= 128:
            out.append((value & 0x7F) | 0x80)
            value >>= 7
        out.append(value)
        return bytes(out)

    def _string(self, s: str) -> bytes:
        raw = s.encode()
        huff = huffman_encode(raw)
        if len(huff) < len(raw):
            return self._int(7, 0x80, len(huff)) + huff
        return self._int(7, 0x00, len(raw)) + raw

    def encode(self, headers, sensitive=()):
        out = bytearray()
        for name, value in headers:
            exact, name_idx = self._find(name, value)
            if name in sensitive:
                # never-indexed literal (grpc-go: authorization etc.)
                out += self._int(4, 0x10, name_idx if name_idx < 62 else 0)
                if not name_idx or name_idx >= 62:
                    out += self._string(name)
                out += self._string(value)
                continue
            if exact:
                out += self._int(7, 0x80, exact)
                continue
            # literal with incremental indexing
            out += self._int(6, 0x40, name_idx)
            if not name_idx:
                out += self._string(name)
            out += self._string(value)
            self.dyn.insert(0, (name, value))
            self._evict()
        return bytes(out)


class GoHpackDecoder:
    """Just enough HPACK decoding to read the plugin's response headers
    (our server emits static-table indexed + plain literals, no huffman,
    no dynamic inserts — but decode all forms for robustness)."""

    def __init__(self):
        self.dyn = []

    def _lookup(self, idx):
        if 1 <= idx <= 61:
            return STATIC_TABLE[idx - 1]
        di = idx - 62
        assert 0 <= di < len(self.dyn), f"bad index {idx}"
        return self.dyn[di]

    @staticmethod
    def _read_int(buf, pos, prefix_bits):
        limit = (1 << prefix_bits) - 1
        val = buf[pos] & limit
        pos += 1
        if val == limit:
            shift = 0
            while True:
                b = buf[pos]
                pos += 1
                val += (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
        return val, pos

    def _read_string(self, buf, pos):
        huff = bool(buf[pos] & 0x80)
        length, pos = self._read_int(buf, pos, 7)
        raw = buf[pos:pos + length]
        pos += length
        if huff:
            raw = huffman_decode(raw)
        return raw.decode(), pos

    def decode(self, block: bytes):
        out = []
        pos = 0
        while pos < len(block):
            b = block[pos]
            if b & 0x80:
                idx, pos = self._read_int(block, pos, 7)
                out.append(self._lookup(idx))
            elif b & 0x40:
                idx, pos = self._read_int(block, pos, 6)
                if idx:
                    name = self._lookup(idx)[0]
                else:
                    pos += 1
                    name, pos = self._read_string(block, pos - 1)
                value, pos = self._read_string(block, pos)
                self.dyn.insert(0, (name, value))
                out.append((name, value))
            elif b & 0x20:
                _, pos = self._read_int(block, pos, 5)  # table size update
            else:
                idx, pos = self._read_int(block, pos, 4)
                if idx:
                    name = self._lookup(idx)[0]
                else:
                    name, pos = self._read_string(block, pos)
                value, pos = self._read_string(block, pos)
                out.append((name, value))
        return out


def _build_huff_decode_map():
    return {(code, bits): sym for sym, (code, bits) in enumerate(_HUFF)}


_HUFF_DECODE = _build_huff_decode_map()


def huffman_decode(data: bytes) -> bytes:
    out = bytearray()
    acc, nbits = 0, 0
    total = len(data) * 8
    consumed = 0
    for byte in data:
        acc = (acc << 8) | byte
        nbits += 8
        while nbits >= 5:  # min code length is 5
            # trailing all-ones (< 8 bits left overall) is EOS padding
            if total - consumed - nbits == 0 and nbits <= 7 and \
                    acc == (1 << nbits) - 1:
                return bytes(out)
            matched = False
            for bits in range(5, min(nbits, 30) + 1):
                code = (acc >> (nbits - bits)) & ((1 << bits) - 1)
                sym = _HUFF_DECODE.get((code, bits))
                if sym is not None and sym < 256:
                    out.append(sym)
                    nbits -= bits
                    consumed += bits
                    acc &= (1 << nbits) - 1
                    matched = True
                    break
            if not matched:
                break
    return bytes(out)


class GrpcGoConn:
    """One client connection behaving like grpc-go's http2Client."""

    USER_AGENT = "grpc-go/1.62.0"

    def __init__(self, unix_path, timeout=10):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.connect(unix_path)
        self.sock.settimeout(timeout)
        self.buf = b""
        self.enc = GoHpackEncoder()
        self.dec = GoHpackDecoder()
        self.next_stream = 1
        self.settings_acked = False
        self.server_settings = {}
        # grpc-go: preface, SETTINGS{MAX_FRAME_SIZE}, conn WINDOW_UPDATE
        self.sock.sendall(
            PREFACE +
            frame(SETTINGS, 0, 0,
                  struct.pack(">HI", SETTINGS_MAX_FRAME_SIZE, 16384)) +
            frame(WINDOW_UPDATE, 0, 0, struct.pack(">I", 983041)))

    # --- low level ---------------------------------------------------------
    def send(self, data):
        self.sock.sendall(data)

    def read_frame(self):
        while len(self.buf) < 9:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise ConnectionError("peer closed")
            self.buf += chunk
        length = int.from_bytes(self.buf[:3], "big")
        ftype, flags = self.buf[3], self.buf[4]
        stream = int.from_bytes(self.buf[5:9], "big") & 0x7FFFFFFF
        while len(self.buf) < 9 + length:
            chunk = self.sock.recv(65536)
            if not chunk:
                raise ConnectionError("peer closed mid-frame")
            self.buf += chunk
        payload = self.buf[9:9 + length]
        self.buf = self.buf[9 + length:]
        return ftype, flags, stream, payload

    def handle_control(self, ftype, flags, stream, payload):
        """The transport-level reactions grpc-go performs automatically.
        Returns True if the frame was a control frame."""
        if ftype == SETTINGS:
            if flags & ACK:
                self.settings_acked = True
            else:
                for i in range(0, len(payload) - 5, 6):
                    sid, val = struct.unpack(">HI", payload[i:i + 6])
                    self.server_settings[sid] = val
                self.send(frame(SETTINGS, ACK, 0))
            return True
        if ftype == PING and not flags & ACK:
            self.send(frame(PING, ACK, 0, payload))
            return True
        return ftype in (PING, WINDOW_UPDATE, PRIORITY)

    # --- RPC surface -------------------------------------------------------
    def request_headers(self, path, timeout_header=None):
        hs = [(":method", "POST"), (":scheme", "http"), (":path", path),
              (":authority", "localhost"),
              ("content-type", "application/grpc"),
              ("user-agent", self.USER_AGENT), ("te", "trailers")]
        if timeout_header:
            hs.append(("grpc-timeout", timeout_header))
        return self.enc.encode(hs)

    def start_stream(self, path, body=b"", end_stream=True,
                     timeout_header=None):
        sid = self.next_stream
        self.next_stream += 2
        self.send(frame(HEADERS, END_HEADERS, sid,
                        self.request_headers(path, timeout_header)))
        self.send(frame(DATA, END_STREAM if end_stream else 0, sid,
                        grpc_frame(body)))
        return sid

    def read_stream_event(self, want_stream, *, bdp_ping_on_data=False):
        """Next (kind, payload) event on want_stream; kind in
        {'headers', 'data', 'trailers', 'rst'}. Control frames handled
        transparently; DATA triggers a window-credit return (and
        optionally a BDP ping), like grpc-go's transport. Events for
        OTHER streams are buffered (grpc-go multiplexes concurrent RPCs
        on one connection), so interleaved Allocate-during-ListAndWatch
        flows demux correctly."""
        pending = getattr(self, "_pending", None)
        if pending is None:
            pending = self._pending = {}
        if pending.get(want_stream):
            return pending[want_stream].pop(0)
        while True:
            ftype, flags, stream, payload = self.read_frame()
            if self.handle_control(ftype, flags, stream, payload):
                continue
            if stream != want_stream:
                ev = self._classify(ftype, flags, stream, payload,
                                    bdp_ping_on_data=False)
                if ev is not None:
                    pending.setdefault(stream, []).append(ev)
                continue
            ev = self._classify(ftype, flags, stream, payload,
                                bdp_ping_on_data=bdp_ping_on_data)
            if ev is not None:
                return ev

    def _classify(self, ftype, flags, stream, payload, *,
                  bdp_ping_on_data):
        if ftype == DATA:
            if payload:
                inc = struct.pack(">I", len(payload))
                self.send(frame(WINDOW_UPDATE, 0, stream, inc))
                self.send(frame(WINDOW_UPDATE, 0, 0, inc))
            if bdp_ping_on_data:
                self.send(frame(PING, 0, 0, BDP_PING))
            return ("data", payload)
        if ftype == HEADERS:
            hs = self.dec.decode(payload)
            kind = "trailers" if any(n == "grpc-status" for n, _ in hs) \
                else "headers"
            return (kind, hs)
        if ftype == RST_STREAM:
            return ("rst", payload)
        return None

    def unary(self, path, body=b"", timeout_header=None, **ev_kw):
        sid = self.start_stream(path, body, timeout_header=timeout_header)
        data = b""
        status = None
        while status is None:
            kind, payload = self.read_stream_event(sid, **ev_kw)
            if kind == "data":
                data += payload
            elif kind == "trailers":
                status = int(dict(payload)["grpc-status"])
            elif kind == "rst":
                raise ConnectionError("stream reset")
        assert status == 0, f"grpc-status {status}"
        assert data[:1] == b"\x00"
        mlen = int.from_bytes(data[1:5], "big")
        return data[5:5 + mlen]

    def cancel(self, sid):
        self.send(frame(RST_STREAM, 0, sid, struct.pack(">I", 0x8)))  # CANCEL

    def wait_ping_ack(self, payload):
        while True:
            ftype, flags, stream, p = self.read_frame()
            if ftype == PING and flags & ACK:
                return p == payload
            self.handle_control(ftype, flags, stream, p)

    def goaway_and_close(self):
        # grpc-go Close(): GOAWAY(last_stream=0, NO_ERROR) then TCP close
        self.send(frame(GOAWAY, 0, 0, struct.pack(">II", 0, 0)))
        self.sock.close()

    def close(self):
        self.sock.close()

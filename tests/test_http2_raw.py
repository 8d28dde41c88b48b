"""Raw-frame HTTP/2 conformance tests against the device plugin's server:
padded HEADERS/DATA and CONTINUATION fragmentation — legal protocol shapes
that grpcio never emits but a future kubelet gRPC stack may."""

import socket
import struct
import subprocess
import time
from pathlib import Path

import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"

HEADERS, DATA, SETTINGS, CONTINUATION, WINDOW_UPDATE = 0x1, 0x0, 0x4, 0x9, 0x8
END_STREAM, ACK, END_HEADERS, PADDED = 0x1, 0x1, 0x4, 0x8


def frame(ftype, flags, stream, payload):
    return struct.pack(">I", len(payload))[1:] + bytes([ftype, flags]) + \
        struct.pack(">I", stream) + payload


def hpack_literal(name: bytes, value: bytes) -> bytes:
    # literal without indexing, new name, no huffman (lengths < 127)
    return bytes([0x00, len(name)]) + name + bytes([len(value)]) + value


def grpc_frame(msg: bytes) -> bytes:
    return b"\x00" + struct.pack(">I", len(msg)) + msg


class RawConn:
    def __init__(self, path):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.connect(path)
        self.sock.settimeout(10)
        self.buf = b""
        self.sock.sendall(PREFACE + frame(SETTINGS, 0, 0, b""))

    def send(self, data):
        self.sock.sendall(data)

    def read_frame(self):
        while len(self.buf) < 9:
            self.buf += self.sock.recv(65536)
        length = int.from_bytes(self.buf[:3], "big")
        ftype, flags = self.buf[3], self.buf[4]
        stream = int.from_bytes(self.buf[5:9], "big") & 0x7FFFFFFF
        while len(self.buf) < 9 + length:
            self.buf += self.sock.recv(65536)
        payload = self.buf[9:9 + length]
        self.buf = self.buf[9 + length:]
        return ftype, flags, stream, payload

    def close(self):
        self.sock.close()


@pytest.fixture
def plugin(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=2)
    sock = str(tmp_path / "amd.sock")
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", sock, "--no-register",
         "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    deadline = time.time() + 10
    while not Path(sock).exists():
        assert time.time() < deadline and proc.poll() is None
        time.sleep(0.05)
    yield sock
    proc.terminate()
    proc.wait(timeout=10)


def request_headers(path: bytes) -> bytes:
    return (hpack_literal(b":method", b"POST") +
            hpack_literal(b":scheme", b"http") +
            hpack_literal(b":path", path) +
            hpack_literal(b":authority", b"x") +
            hpack_literal(b"content-type", b"application/grpc") +
            hpack_literal(b"te", b"trailers"))


def run_unary(conn, path, body, *, pad_headers=False, fragment=False,
              pad_data=False):
    block = request_headers(path)
    if fragment:
        mid = len(block) // 2
        flags = PADDED if pad_headers else 0
        first = block[:mid]
        if pad_headers:
            first = bytes([3]) + first + b"\x00" * 3
        conn.send(frame(HEADERS, flags, 1, first))
        conn.send(frame(CONTINUATION, END_HEADERS, 1, block[mid:]))
    else:
        flags = END_HEADERS | (PADDED if pad_headers else 0)
        payload = block
        if pad_headers:
            payload = bytes([4]) + block + b"\x00" * 4
        conn.send(frame(HEADERS, flags, 1, payload))
    data = grpc_frame(body)
    if pad_data:
        conn.send(frame(DATA, END_STREAM | PADDED, 1,
                        bytes([5]) + data + b"\x00" * 5))
    else:
        conn.send(frame(DATA, END_STREAM, 1, data))

    resp_data = b""
    grpc_status = None
    while grpc_status is None:
        ftype, flags, stream, payload = conn.read_frame()
        if ftype == SETTINGS and not flags & ACK:
            conn.send(frame(SETTINGS, ACK, 0, b""))
        elif ftype == DATA and stream == 1:
            resp_data += payload
        elif ftype == HEADERS and stream == 1:
            # crude scan for the literal "grpc-status" name bytes + value
            if b"grpc-status" in payload or flags & END_STREAM:
                # served without huffman by our encoder: find digit value
                idx = payload.find(b"grpc-status")
                assert idx >= 0
                vlen = payload[idx + len(b"grpc-status")]
                val = payload[idx + len(b"grpc-status") + 1:
                              idx + len(b"grpc-status") + 1 + vlen]
                grpc_status = int(val or b"-1")
    assert grpc_status == 0, grpc_status
    assert resp_data[:1] == b"\x00"
    mlen = int.from_bytes(resp_data[1:5], "big")
    return resp_data[5:5 + mlen]


def test_padded_headers_and_data(plugin):
    conn = RawConn(plugin)
    try:
        msg = run_unary(conn, b"/v1beta1.DevicePlugin/GetDevicePluginOptions",
                        b"", pad_headers=True, pad_data=True)
        opts = pb.decode_options(msg)
        assert opts["get_preferred_allocation_available"] is True
    finally:
        conn.close()


def test_continuation_fragmented_headers(plugin):
    conn = RawConn(plugin)
    try:
        msg = run_unary(conn, b"/v1beta1.DevicePlugin/GetDevicePluginOptions",
                        b"", fragment=True)
        opts = pb.decode_options(msg)
        assert opts["pre_start_required"] is False
    finally:
        conn.close()


def test_unknown_method_gets_unimplemented(plugin):
    conn = RawConn(plugin)
    try:
        conn.send(frame(HEADERS, END_HEADERS, 1,
                        request_headers(b"/v1beta1.DevicePlugin/Nope")))
        conn.send(frame(DATA, END_STREAM, 1, grpc_frame(b"")))
        status = None
        while status is None:
            ftype, flags, stream, payload = conn.read_frame()
            if ftype == SETTINGS and not flags & ACK:
                conn.send(frame(SETTINGS, ACK, 0, b""))
            elif ftype == HEADERS and b"grpc-status" in payload:
                idx = payload.find(b"grpc-status")
                vlen = payload[idx + 11]
                status = int(payload[idx + 12: idx + 12 + vlen])
        assert status == 12  # UNIMPLEMENTED
    finally:
        conn.close()


def test_unary_response_through_tiny_flow_control_window(plugin):
    """A unary response larger than the peer's stream window must not
    deadlock: the server's frame-reader has to keep consuming
    WINDOW_UPDATE frames while the (worker-thread) unary handler waits
    for window credit. Regression test for the reader-thread-sends-unary
    design (grpc_transport.cc) — under the old inline dispatch this
    stalled 60 s and failed the RPC."""
    conn = RawConn(plugin)
    try:
        # shrink every stream's send window to 64 bytes
        conn.send(frame(SETTINGS, 0, 0, struct.pack(">HI", 4, 64)))
        ids = ["amdgpu-1a2b3c4d5e6f0000", "amdgpu-1a2b3c4d5e6f0001"]
        conn.send(frame(HEADERS, END_HEADERS, 1,
                        request_headers(b"/v1beta1.DevicePlugin/Allocate")))
        conn.send(frame(DATA, END_STREAM, 1,
                        grpc_frame(pb.encode_allocate_request([ids]))))
        t0 = time.time()
        resp_data = b""
        done = False
        while not done:
            ftype, flags, stream, payload = conn.read_frame()
            if ftype == SETTINGS and not flags & ACK:
                conn.send(frame(SETTINGS, ACK, 0, b""))
            elif ftype == DATA and stream == 1:
                assert len(payload) <= 64, "server ignored the stream window"
                resp_data += payload
                # drip window credit back: stream + connection level
                inc = struct.pack(">I", max(len(payload), 1))
                conn.send(frame(WINDOW_UPDATE, 0, 1, inc))
                conn.send(frame(WINDOW_UPDATE, 0, 0, inc))
            elif ftype == HEADERS and stream == 1 and \
                    b"grpc-status" in payload:
                idx = payload.find(b"grpc-status")
                vlen = payload[idx + 11]
                assert int(payload[idx + 12: idx + 12 + vlen]) == 0
                done = True
        assert time.time() - t0 < 20, "flow-control deadlock"
        assert resp_data[:1] == b"\x00"
        mlen = int.from_bytes(resp_data[1:5], "big")
        assert len(resp_data) >= 5 + mlen > 64  # actually exercised chunking
        resp = pb.decode_allocate_response(resp_data[5:5 + mlen])
        env = resp[0]["envs"]
        assert env["K3SAMD_VISIBLE_DEVICES"] == ",".join(ids)
    finally:
        conn.close()

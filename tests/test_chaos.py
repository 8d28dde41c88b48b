"""Chaos tests: hostile/broken byte streams against the gRPC server must
never take the device plugin down (the SIGPIPE churn crash was exactly
this class of bug)."""

import socket
import struct
import subprocess
import time
from pathlib import Path

import grpc
import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"
PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"

IDENT = lambda b: b  # noqa: E731


@pytest.fixture
def plugin(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=1)
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
    yield sock, proc
    proc.terminate()
    proc.wait(timeout=10)


def raw(sock_path, payload, linger=0.05):
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.connect(sock_path)
    if payload:
        s.sendall(payload)
    time.sleep(linger)
    s.close()


def frame(ftype, flags, stream, payload):
    return struct.pack(">I", len(payload))[1:] + bytes([ftype, flags]) + \
        struct.pack(">I", stream) + payload


def assert_alive(sock_path, proc):
    assert proc.poll() is None, proc.stderr.read()[-500:]
    ch = grpc.insecure_channel(f"unix:{sock_path}")
    opt = ch.unary_unary("/v1beta1.DevicePlugin/GetDevicePluginOptions",
                         request_serializer=IDENT,
                         response_deserializer=IDENT)
    opts = pb.decode_options(opt(b"", timeout=10))
    assert opts["get_preferred_allocation_available"] is True
    ch.close()


def test_survives_hostile_streams(plugin):
    sock, proc = plugin
    cases = [
        b"",                                   # connect + close
        b"\x00" * 64,                          # garbage, no preface
        b"GET / HTTP/1.1\r\n\r\n",             # wrong protocol
        PREFACE,                               # preface then hangup
        PREFACE + b"\xff" * 32,                # bogus frames
        PREFACE + frame(0x1, 0x4, 1, b"\xff\xfe\xfd"),   # invalid HPACK
        PREFACE + frame(0x1, 0x4, 1, b"")[:7],           # truncated header
        PREFACE + b"\xff\xff\xff\x00\x00\x00\x00\x00\x01",  # 16MB len claim
        PREFACE + frame(0x6, 0x0, 0, b"12345678") * 50,  # ping flood
        PREFACE + frame(0x9, 0x4, 1, b"x"),    # CONTINUATION w/o HEADERS
    ]
    for payload in cases:
        for _ in range(3):
            raw(sock, payload)
    assert_alive(sock, proc)


def test_survives_disconnect_during_response(plugin):
    """Peer disappears right after sending a valid request: the server's
    response writes hit a closed socket (the EPIPE path)."""
    sock, proc = plugin
    from test_http2_raw import request_headers, grpc_frame
    for _ in range(10):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.connect(sock)
        s.sendall(PREFACE + frame(0x4, 0, 0, b"") +
                  frame(0x1, 0x4, 1,
                        request_headers(
                            b"/v1beta1.DevicePlugin/GetDevicePluginOptions")) +
                  frame(0x0, 0x1, 1, grpc_frame(b"")))
        s.close()  # gone before reading the response
    time.sleep(0.3)
    assert_alive(sock, proc)


def test_survives_many_half_open(plugin):
    sock, proc = plugin
    socks = []
    for _ in range(30):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.connect(sock)
        socks.append(s)
    time.sleep(0.2)
    for s in socks:
        s.close()
    assert_alive(sock, proc)


def test_structured_frame_fuzz(plugin):
    """Seeded random VALID-ish frame interleavings (settings, pings,
    window updates, priority, padded headers/data fragments, rst,
    unknown types) — the server must never crash and must still serve a
    clean connection afterwards."""
    import random
    import struct as struct_mod
    from grpcgo_wire import (GoHpackEncoder, PREFACE, frame, grpc_frame,
                             SETTINGS, PING, WINDOW_UPDATE, HEADERS, DATA,
                             RST_STREAM, END_HEADERS, END_STREAM)
    sock_path, _proc = plugin
    rng = random.Random(1234)
    for trial in range(8):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.connect(sock_path)
        s.settimeout(5)
        out = bytearray(PREFACE + frame(SETTINGS, 0, 0))
        enc = GoHpackEncoder()
        sid = 1
        for _ in range(rng.randint(3, 25)):
            k = rng.randint(0, 7)
            if k == 0:
                out += frame(SETTINGS, 0, 0,
                             struct_mod.pack(">HI", rng.randint(1, 6),
                                             rng.randint(0, 1 << 20)))
            elif k == 1:
                out += frame(PING, 0, 0, bytes(rng.randrange(256)
                                               for _ in range(8)))
            elif k == 2:
                out += frame(WINDOW_UPDATE, 0, rng.choice([0, sid]),
                             struct_mod.pack(">I", rng.randint(1, 1 << 20)))
            elif k == 3:
                block = enc.encode([(":method", "POST"), (":scheme", "http"),
                                    (":path",
                                     "/v1beta1.DevicePlugin/GetDevicePluginOptions"),
                                    (":authority", "x"),
                                    ("content-type", "application/grpc"),
                                    ("te", "trailers")])
                out += frame(HEADERS, END_HEADERS, sid, block)
                out += frame(DATA, END_STREAM, sid, grpc_frame(b""))
                sid += 2
            elif k == 4:
                out += frame(RST_STREAM, 0, max(1, sid - 2),
                             struct_mod.pack(">I", rng.randint(0, 13)))
            elif k == 5:
                out += frame(2, 0, sid, bytes(5))  # PRIORITY
            elif k == 6:
                out += frame(rng.randint(10, 200), rng.randrange(256),
                             rng.choice([0, sid]),
                             bytes(rng.randrange(64)))  # unknown type
            else:
                out += frame(DATA, 0, 999, b"orphan")  # unknown stream
        try:
            s.sendall(bytes(out))
            s.recv(4096)
        except OSError:
            pass
        s.close()
    # the plugin survived all trials and serves a clean connection
    from grpcgo_wire import GrpcGoConn
    import pb_v1beta1 as pb
    c = GrpcGoConn(sock_path)
    assert pb.decode_options(
        c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions"))
    c.close()

"""grpc-go wire-conformance suite for the device plugin's gRPC stack.

kubelet is grpc-go — the hand-rolled HTTP/2 server's only real client —
and grpc-go's frame behavior differs from the grpc-C-core interop tests
in tests/test_deviceplugin.py. These tests replay grpc-go's actual
connection behaviors (see tests/grpcgo_wire.py for the catalog) against
the live plugin binary, covering the registration + serving flow the
reference relies on (/root/reference/README.md:105-126).
"""

import json
import socket
import struct
import subprocess
import threading
import time
from pathlib import Path

import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree
from grpcgo_wire import (
    ACK, BDP_PING, DATA, END_HEADERS, END_STREAM, GOAWAY, GrpcGoConn,
    GoHpackDecoder, GoHpackEncoder, HEADERS, KEEPALIVE_PING, PING, PREFACE,
    RST_STREAM, SETTINGS, SETTINGS_MAX_FRAME_SIZE, WINDOW_UPDATE, frame,
    grpc_frame,
)

REPO = Path(__file__).resolve().parent.parent
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"

DEFAULT_CFG = """\
version: v1
sharing:
  timeSlicing:
    resources:
    - name: amd.com/gpu
      replicas: {replicas}
"""


def start_plugin(tmp_path, n_gpus=2, replicas=1, kubelet_sock=None):
    root = build_tree(tmp_path / "sys", n_gpus=n_gpus)
    cfg = tmp_path / "config.yaml"
    cfg.write_text(DEFAULT_CFG.format(replicas=replicas))
    sock = str(tmp_path / "amd.sock")
    argv = [str(PLUGIN), "--config", str(cfg), "--plugin-sock", sock,
            "--health-poll-ms", "0"]
    if kubelet_sock:
        argv += ["--kubelet-sock", kubelet_sock]
    else:
        argv += ["--no-register"]
    proc = subprocess.Popen(argv, env={"K3SAMD_SYSFS_ROOT": str(root)},
                            stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    deadline = time.time() + 10
    while not Path(sock).exists():
        assert time.time() < deadline and proc.poll() is None
        time.sleep(0.05)
    return proc, sock


@pytest.fixture
def plugin(tmp_path):
    proc, sock = start_plugin(tmp_path)
    yield sock
    proc.terminate()
    proc.wait(timeout=10)


def test_connection_dance_and_settings_ack(plugin):
    """grpc-go expects: server SETTINGS arrives, client's SETTINGS gets
    ACKed. A server that never ACKs stalls the transport's handshake."""
    c = GrpcGoConn(plugin)
    try:
        msg = c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions")
        opts = pb.decode_options(msg)
        assert opts["get_preferred_allocation_available"] is True
        assert c.settings_acked, "server never ACKed our SETTINGS"
    finally:
        c.close()


def test_dynamic_table_reuse_across_rpcs(plugin):
    """grpc-go HPACK-indexes request headers incrementally: the 2nd/3rd
    RPC's header block is almost entirely dynamic-table references. A
    server whose decoder forgets inserts mis-parses :path and fails."""
    c = GrpcGoConn(plugin)
    try:
        for _ in range(3):
            msg = c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions")
            assert pb.decode_options(msg)["get_preferred_allocation_available"]
        # the encoder really did collapse to indexed fields
        block = c.request_headers("/v1beta1.DevicePlugin/GetDevicePluginOptions")
        assert len(block) < 10, f"header block unexpectedly large: {len(block)}"
    finally:
        c.close()


def test_huffman_coded_headers(plugin):
    """All grpc-go string literals are huffman-coded when shorter — the
    very first request already carries huffman :path/user-agent."""
    c = GrpcGoConn(plugin)
    try:
        # sanity (on a throwaway encoder so the live one stays in sync):
        # the first block contains no raw ASCII of the path — it is
        # huffman-coded, so the server must actually decode huffman
        block = GoHpackEncoder().encode(
            [(":path", "/v1beta1.DevicePlugin/Allocate")])
        assert b"/v1beta1" not in block
        ids = ["amdgpu-1a2b3c4d5e6f0000"]
        resp = pb.decode_allocate_response(
            c.unary("/v1beta1.DevicePlugin/Allocate",
                    pb.encode_allocate_request([ids])))
        assert resp[0]["envs"]["K3SAMD_VISIBLE_DEVICES"] == ids[0]
    finally:
        c.close()


def test_bdp_ping_echoed_mid_stream(plugin):
    """grpc-go sends its BDP probe PING right after response DATA and
    requires the ACK to carry the identical payload; a wrong echo stalls
    window growth for the connection's lifetime."""
    c = GrpcGoConn(plugin)
    try:
        sid = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        # response headers, then the first device-list DATA
        kind, _ = c.read_stream_event(sid)
        assert kind == "headers"
        kind, payload = c.read_stream_event(sid, bdp_ping_on_data=True)
        assert kind == "data" and payload
        assert c.wait_ping_ack(BDP_PING), "BDP ping payload not echoed"
        # keepalive ping as well
        c.send(frame(PING, 0, 0, KEEPALIVE_PING))
        assert c.wait_ping_ack(KEEPALIVE_PING)
    finally:
        c.close()


def test_rst_stream_mid_listandwatch_then_new_rpc(plugin):
    """kubelet restarting its ListAndWatch: RST_STREAM(CANCEL) on the
    live stream, then a fresh RPC on the SAME connection must work and
    the cancelled stream's thread must stop pushing frames."""
    c = GrpcGoConn(plugin)
    try:
        sid = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(sid)          # headers
        kind, data = c.read_stream_event(sid)       # first update
        assert kind == "data"
        c.cancel(sid)
        # same connection, next odd stream id — grpc-go reuses transports
        msg = c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions")
        assert pb.decode_options(msg)["pre_start_required"] is False
        # restart ListAndWatch (the kubelet-restart pattern)
        sid2 = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(sid2)
        assert kind == "headers"
        kind, data = c.read_stream_event(sid2)
        assert kind == "data"
        devs = pb.decode_list_and_watch(data[5:])
        assert len(devs) == 2
    finally:
        c.close()


def test_grpc_timeout_header_tolerated(plugin):
    """kubelet's calls carry grpc-timeout metadata (e.g. '10S'); the
    server must parse past it (it is huffman + incrementally indexed)."""
    c = GrpcGoConn(plugin)
    try:
        msg = c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions",
                      timeout_header="10S")
        assert pb.decode_options(msg)
        # and with a different unit + value (fresh dynamic-table entry)
        msg = c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions",
                      timeout_header="500m")
        assert pb.decode_options(msg)
    finally:
        c.close()


def test_goaway_close_leaves_plugin_serving(tmp_path):
    """grpc-go sends GOAWAY(NO_ERROR) on transport close; the plugin must
    drop the connection cleanly and keep serving new ones."""
    proc, sock = start_plugin(tmp_path)
    try:
        c = GrpcGoConn(sock)
        assert pb.decode_options(
            c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions"))
        c.goaway_and_close()
        time.sleep(0.1)
        assert proc.poll() is None, "plugin died on GOAWAY"
        c2 = GrpcGoConn(sock)
        assert pb.decode_options(
            c2.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions"))
        c2.close()
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_max_frame_size_respected_on_large_response(tmp_path):
    """grpc-go advertises SETTINGS_MAX_FRAME_SIZE=16384 and treats larger
    DATA frames as a connection error. Drive a ListAndWatch response past
    16 KiB (8 GPUs x 64 replicas = 512 device entries) and check every
    frame's size."""
    proc, sock = start_plugin(tmp_path, n_gpus=8, replicas=64)
    try:
        c = GrpcGoConn(sock)
        sid = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(sid)
        assert kind == "headers"
        data = b""
        while True:
            ftype, flags, stream, payload = c.read_frame()
            if c.handle_control(ftype, flags, stream, payload):
                continue
            if ftype == DATA and stream == sid:
                assert len(payload) <= 16384, \
                    f"DATA frame {len(payload)} exceeds MAX_FRAME_SIZE"
                data += payload
                inc = struct.pack(">I", len(payload))
                c.send(frame(WINDOW_UPDATE, 0, sid, inc))
                c.send(frame(WINDOW_UPDATE, 0, 0, inc))
                mlen = int.from_bytes(data[1:5], "big")
                if len(data) >= 5 + mlen:
                    break
        devs = pb.decode_list_and_watch(data[5:5 + int.from_bytes(
            data[1:5], "big")])
        assert len(devs) == 512
        assert len(data) > 16384, "response did not actually exceed one frame"
        c.close()
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_zero_initial_window_then_open(plugin):
    """grpc-go's dynamic windowing can shrink the stream window; the edge
    is INITIAL_WINDOW_SIZE=0 — new streams start frozen and only a later
    WINDOW_UPDATE releases the response."""
    c = GrpcGoConn(plugin)
    try:
        c.send(frame(SETTINGS, 0, 0,
                     struct.pack(">HI", 0x4, 0)))  # INITIAL_WINDOW_SIZE=0
        sid = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(sid)
        assert kind == "headers"  # headers are not flow-controlled
        # no DATA may arrive while the window is 0
        c.sock.settimeout(1.0)
        got_data = False
        try:
            while True:
                ftype, flags, stream, payload = c.read_frame()
                if c.handle_control(ftype, flags, stream, payload):
                    continue
                if ftype == DATA and stream == sid and payload:
                    got_data = True
                    break
        except (socket.timeout, TimeoutError):
            pass
        assert not got_data, "server sent DATA into a zero window"
        # open the stream window; the device list must now flow
        c.sock.settimeout(10.0)
        c.send(frame(WINDOW_UPDATE, 0, sid, struct.pack(">I", 1 << 20)))
        kind, payload = c.read_stream_event(sid)
        assert kind == "data" and payload
    finally:
        c.close()


class FakeGrpcGoKubelet:
    """A Registration server speaking with grpc-go server mannerisms:
    SETTINGS{MAX_FRAME_SIZE} + connection WINDOW_UPDATE on accept, a BDP
    PING before responding, huffman + incrementally-indexed response
    headers and trailers. The plugin's *client* must interoperate."""

    def __init__(self, sock_path):
        self.path = sock_path
        self.requests = []
        self.got_ping_ack = threading.Event()
        self.done = threading.Event()
        self.srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        Path(sock_path).unlink(missing_ok=True)
        self.srv.bind(sock_path)
        self.srv.listen(1)
        self.thread = threading.Thread(target=self._serve, daemon=True)
        self.thread.start()

    def _serve(self):
        try:
            conn, _ = self.srv.accept()
        except OSError:
            return
        conn.settimeout(10)
        enc = GoHpackEncoder()
        dec = GoHpackDecoder()
        buf = b""

        def read_frame():
            nonlocal buf
            while len(buf) < 9:
                buf += conn.recv(65536)
            ln = int.from_bytes(buf[:3], "big")
            ft, fl = buf[3], buf[4]
            st = int.from_bytes(buf[5:9], "big") & 0x7FFFFFFF
            while len(buf) < 9 + ln:
                buf += conn.recv(65536)
            pl = buf[9:9 + ln]
            buf = buf[9 + ln:]
            return ft, fl, st, pl

        try:
            # read client preface
            need = len(PREFACE)
            pre = b""
            while len(pre) < need:
                pre += conn.recv(need - len(pre))
            assert pre == PREFACE, pre
            # grpc-go server greeting
            conn.sendall(
                frame(SETTINGS, 0, 0,
                      struct.pack(">HI", SETTINGS_MAX_FRAME_SIZE, 16384)) +
                frame(WINDOW_UPDATE, 0, 0, struct.pack(">I", 983041)))
            req_data = b""
            stream_id = None
            while True:
                ft, fl, st, pl = read_frame()
                if ft == SETTINGS and not fl & ACK:
                    conn.sendall(frame(SETTINGS, ACK, 0))
                elif ft == HEADERS:
                    stream_id = st
                elif ft == DATA and st == stream_id:
                    req_data += pl
                    if fl & END_STREAM:
                        break
                elif ft == PING and not fl & ACK:
                    conn.sendall(frame(PING, ACK, 0, pl))
            mlen = int.from_bytes(req_data[1:5], "big")
            self.requests.append(req_data[5:5 + mlen])
            # BDP probe before answering — the client MUST ack it
            conn.sendall(frame(PING, 0, 0, BDP_PING))
            # response: huffman/indexed headers, empty message, trailers
            conn.sendall(frame(
                HEADERS, END_HEADERS, stream_id,
                enc.encode([(":status", "200"),
                            ("content-type", "application/grpc")])))
            conn.sendall(frame(DATA, 0, stream_id, grpc_frame(b"")))
            conn.sendall(frame(
                HEADERS, END_HEADERS | END_STREAM, stream_id,
                enc.encode([("grpc-status", "0"), ("grpc-message", "")])))
            # confirm the BDP ping came back before the client hung up
            deadline = time.time() + 5
            while time.time() < deadline:
                ft, fl, st, pl = read_frame()
                if ft == PING and fl & ACK and pl == BDP_PING:
                    self.got_ping_ack.set()
                    break
        except Exception:
            pass
        finally:
            self.done.set()
            conn.close()

    def stop(self):
        self.srv.close()


def test_plugin_client_registers_with_grpcgo_server(tmp_path):
    """Reverse direction: the plugin's Register client against a server
    with grpc-go mannerisms (BDP ping mid-RPC, huffman + dynamic-table
    response headers). Registration must succeed and the plugin must
    keep serving."""
    kubelet_sock = str(tmp_path / "kubelet.sock")
    fake = FakeGrpcGoKubelet(kubelet_sock)
    proc, sock = start_plugin(tmp_path, kubelet_sock=kubelet_sock)
    try:
        assert fake.done.wait(10), "registration RPC never completed"
        assert fake.requests, "no Register request received"
        req = pb.decode_register_request(fake.requests[0])
        assert req["resource_name"] == "amd.com/gpu"
        assert fake.got_ping_ack.is_set(), "client never ACKed the BDP ping"
        # plugin survived and serves (did not treat the dance as an error)
        time.sleep(0.2)
        assert proc.poll() is None
        c = GrpcGoConn(sock)
        assert pb.decode_options(
            c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions"))
        c.close()
    finally:
        fake.stop()
        proc.terminate()
        proc.wait(timeout=10)


def test_interleaved_allocate_during_listandwatch(plugin):
    """kubelet's real concurrency pattern: ListAndWatch stays open while
    Allocate RPCs multiplex on the SAME connection. Events must demux by
    stream id — the Allocate response arrives while LW frames may be
    in flight, and the LW stream keeps working afterwards."""
    c = GrpcGoConn(plugin)
    try:
        lw = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(lw)
        assert kind == "headers"
        kind, first = c.read_stream_event(lw)
        assert kind == "data"
        devs = pb.decode_list_and_watch(first[5:])
        # now, with LW open, run THREE Allocates on the same connection
        for d in devs[:2] + [devs[0]]:
            resp = pb.decode_allocate_response(
                c.unary("/v1beta1.DevicePlugin/Allocate",
                        pb.encode_allocate_request([[d["id"]]])))
            assert d["id"] in resp[0]["envs"]["K3SAMD_VISIBLE_DEVICES"]
        # LW must still be alive: cancel + reopen works on this conn
        c.cancel(lw)
        lw2 = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        kind, _ = c.read_stream_event(lw2)
        assert kind == "headers"
        kind, data = c.read_stream_event(lw2)
        assert len(pb.decode_list_and_watch(data[5:])) == len(devs)
    finally:
        c.close()


def test_bytewise_tcp_fragmentation(plugin):
    """TCP gives no framing guarantees: deliver an entire RPC one byte
    per write. The server's frame reassembly must be position-agnostic."""
    import socket as socket_mod
    raw = socket_mod.socket(socket_mod.AF_UNIX, socket_mod.SOCK_STREAM)
    raw.connect(plugin)
    raw.settimeout(15)
    enc = GoHpackEncoder()
    block = enc.encode([(":method", "POST"), (":scheme", "http"),
                        (":path", "/v1beta1.DevicePlugin/GetDevicePluginOptions"),
                        (":authority", "x"),
                        ("content-type", "application/grpc"),
                        ("te", "trailers")])
    wire = (PREFACE + frame(SETTINGS, 0, 0) +
            frame(HEADERS, END_HEADERS, 1, block) +
            frame(DATA, END_STREAM, 1, grpc_frame(b"")))
    for i in range(len(wire)):
        raw.sendall(wire[i:i + 1])
    # read until trailers with grpc-status 0
    buf = b""
    status = None
    while status is None:
        chunk = raw.recv(65536)
        assert chunk, "server closed on fragmented input"
        buf += chunk
        while len(buf) >= 9:
            ln = int.from_bytes(buf[:3], "big")
            if len(buf) < 9 + ln:
                break
            ftype, flags = buf[3], buf[4]
            payload = buf[9:9 + ln]
            buf = buf[9 + ln:]
            if ftype == SETTINGS and not flags & ACK:
                raw.sendall(frame(SETTINGS, ACK, 0))
            elif ftype == HEADERS and b"grpc-status" in payload:
                idx = payload.find(b"grpc-status")
                vlen = payload[idx + 11]
                status = int(payload[idx + 12: idx + 12 + vlen])
    assert status == 0
    raw.close()


def test_conformance_dance_under_tsan(tmp_path):
    """The grpc-go connection dance against the ThreadSanitizer build of
    the plugin — the hand-rolled HTTP/2 threading under its strictest
    client, with the race detector on."""
    import shutil
    tsan = PLUGIN.parent / "k3samd-device-plugin-tsan"
    if not tsan.exists():
        subprocess.run(["make", "-C", str(REPO / "native"), "tsan"],
                       check=True, capture_output=True, timeout=600)
    root = build_tree(tmp_path / "sys", n_gpus=2)
    cfg = tmp_path / "config.yaml"
    cfg.write_text(DEFAULT_CFG.format(replicas=2))
    sock = str(tmp_path / "amd.sock")
    metrics_sock = str(tmp_path / "metrics.sock")
    proc = subprocess.Popen(
        [str(tsan), "--config", str(cfg), "--plugin-sock", sock,
         "--no-register", "--health-poll-ms", "50",
         "--metrics-addr", f"unix:{metrics_sock}"],
        env={"K3SAMD_SYSFS_ROOT": str(root),
             "TSAN_OPTIONS": "exitcode=66 halt_on_error=0"},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 15
        while not Path(sock).exists():
            assert time.time() < deadline and proc.poll() is None
            time.sleep(0.05)
        c = GrpcGoConn(sock)
        lw = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        assert c.read_stream_event(lw)[0] == "headers"
        assert c.read_stream_event(lw, bdp_ping_on_data=True)[0] == "data"
        assert c.wait_ping_ack(BDP_PING)
        for _ in range(3):
            c.unary("/v1beta1.DevicePlugin/GetDevicePluginOptions")
            # concurrent metrics scrapes + a silent client while RPCs run
            # (per-connection metric threads under the race detector)
            ms = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            ms.connect(metrics_sock)
            ms.sendall(b"GET /metrics HTTP/1.0\r\n\r\n")
            while ms.recv(65536):
                pass
            ms.close()
        silent = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        silent.connect(metrics_sock)  # never sends; 2 s timeout path
        c.cancel(lw)
        c.goaway_and_close()
        time.sleep(0.3)
        silent.close()
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait()
    err = proc.stderr.read().decode(errors="replace")
    assert "WARNING: ThreadSanitizer" not in err, err[-3000:]
    assert proc.returncode != 66, "TSan reported races"


class FlakyGrpcGoKubelet:
    """First Register gets a grpc-go TRAILERS-ONLY error response (the
    wire shape grpc-go uses for immediate errors: one HEADERS frame with
    :status + grpc-status, END_STREAM, no DATA); the second connection
    succeeds. Exercises the plugin client's trailers-only parsing AND
    its retry loop."""

    def __init__(self, sock_path):
        self.path = sock_path
        self.register_ok = threading.Event()
        self.attempts = 0
        self.srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        Path(sock_path).unlink(missing_ok=True)
        self.srv.bind(sock_path)
        self.srv.listen(4)
        self.thread = threading.Thread(target=self._serve, daemon=True)
        self.thread.start()

    def _one(self, conn, fail):
        conn.settimeout(10)
        enc = GoHpackEncoder()
        buf = b""

        def read_frame():
            nonlocal buf
            while len(buf) < 9:
                buf += conn.recv(65536)
            ln = int.from_bytes(buf[:3], "big")
            ft, fl = buf[3], buf[4]
            st = int.from_bytes(buf[5:9], "big") & 0x7FFFFFFF
            while len(buf) < 9 + ln:
                buf += conn.recv(65536)
            pl = buf[9:9 + ln]
            buf = buf[9 + ln:]
            return ft, fl, st, pl

        pre = b""
        while len(pre) < len(PREFACE):
            pre += conn.recv(len(PREFACE) - len(pre))
        conn.sendall(frame(SETTINGS, 0, 0))
        sid = None
        while True:
            ft, fl, st, pl = read_frame()
            if ft == SETTINGS and not fl & ACK:
                conn.sendall(frame(SETTINGS, ACK, 0))
            elif ft == HEADERS:
                sid = st
            elif ft == DATA and st == sid and fl & END_STREAM:
                break
        if fail:
            # trailers-only error: single HEADERS, END_STREAM, no body
            conn.sendall(frame(
                HEADERS, END_HEADERS | END_STREAM, sid,
                enc.encode([(":status", "200"),
                            ("content-type", "application/grpc"),
                            ("grpc-status", "14"),
                            ("grpc-message", "registry not ready")])))
        else:
            conn.sendall(frame(HEADERS, END_HEADERS, sid,
                               enc.encode([(":status", "200"),
                                           ("content-type",
                                            "application/grpc")])))
            conn.sendall(frame(DATA, 0, sid, grpc_frame(b"")))
            conn.sendall(frame(HEADERS, END_HEADERS | END_STREAM, sid,
                               enc.encode([("grpc-status", "0"),
                                           ("grpc-message", "")])))
            self.register_ok.set()

    def _serve(self):
        while not self.register_ok.is_set():
            try:
                conn, _ = self.srv.accept()
            except OSError:
                return
            self.attempts += 1
            try:
                self._one(conn, fail=self.attempts == 1)
            except Exception:
                pass
            finally:
                conn.close()

    def stop(self):
        self.srv.close()


def test_trailers_only_error_then_retry_succeeds(tmp_path):
    kubelet_sock = str(tmp_path / "kubelet.sock")
    fake = FlakyGrpcGoKubelet(kubelet_sock)
    root = build_tree(tmp_path / "sys", n_gpus=1)
    cfg = tmp_path / "config.yaml"
    cfg.write_text(DEFAULT_CFG.format(replicas=1))
    sock = str(tmp_path / "amd.sock")
    proc = subprocess.Popen(
        [str(PLUGIN), "--config", str(cfg), "--plugin-sock", sock,
         "--kubelet-sock", kubelet_sock, "--register-retries", "5",
         "--register-backoff-ms", "200", "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        assert fake.register_ok.wait(15), "second Register never succeeded"
        assert fake.attempts == 2
        time.sleep(0.2)
        assert proc.poll() is None  # serving after the retry
    finally:
        fake.stop()
        proc.terminate()
        proc.wait(timeout=10)
        err = proc.stderr.read().decode()
    assert "grpc=14" in err  # the trailers-only error surfaced in the log


def test_conformance_dance_under_asan(tmp_path):
    """The same grpc-go dance against the AddressSanitizer build —
    memory safety of the whole serving path (HPACK huffman/dynamic
    table, frame reassembly, flow control) under the strictest client."""
    asan = PLUGIN.parent / "k3samd-device-plugin-asan"
    if not asan.exists():
        subprocess.run(["make", "-C", str(REPO / "native"), "asan-plugin"],
                       check=True, capture_output=True, timeout=600)
    root = build_tree(tmp_path / "sys", n_gpus=2)
    cfg = tmp_path / "config.yaml"
    cfg.write_text(DEFAULT_CFG.format(replicas=2))
    sock = str(tmp_path / "amd.sock")
    proc = subprocess.Popen(
        [str(asan), "--config", str(cfg), "--plugin-sock", sock,
         "--no-register", "--health-poll-ms", "50"],
        env={"K3SAMD_SYSFS_ROOT": str(root),
             "ASAN_OPTIONS": "exitcode=66 detect_leaks=0"},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 15
        while not Path(sock).exists():
            assert time.time() < deadline and proc.poll() is None
            time.sleep(0.05)
        c = GrpcGoConn(sock)
        lw = c.start_stream("/v1beta1.DevicePlugin/ListAndWatch")
        assert c.read_stream_event(lw)[0] == "headers"
        kind, data = c.read_stream_event(lw, bdp_ping_on_data=True)
        assert kind == "data"
        assert c.wait_ping_ack(BDP_PING)
        devs = pb.decode_list_and_watch(data[5:])
        resp = pb.decode_allocate_response(
            c.unary("/v1beta1.DevicePlugin/Allocate",
                    pb.encode_allocate_request([[devs[0]["id"]]])))
        assert resp[0]["envs"]["K3SAMD_RENDER_MINORS"]
        c.cancel(lw)
        c.goaway_and_close()
        time.sleep(0.3)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait()
    err = proc.stderr.read().decode(errors="replace")
    assert "AddressSanitizer" not in err, err[-3000:]
    assert proc.returncode != 66, "ASan reported memory errors"

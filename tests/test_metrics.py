"""Prometheus metrics endpoint tests (device plugin --metrics-addr)."""

import socket
import subprocess
import time
from pathlib import Path

import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"

IDENT = lambda b: b  # noqa: E731


def scrape(metrics_sock):
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.connect(metrics_sock)
    s.sendall(b"GET /metrics HTTP/1.0\r\n\r\n")
    data = b""
    while chunk := s.recv(65536):
        data += chunk
    s.close()
    head, body = data.split(b"\r\n\r\n", 1)
    assert b"200 OK" in head
    return None, body.decode()


def parse_metrics(body: str) -> dict:
    out = {}
    for line in body.splitlines():
        if line.startswith("#") or not line.strip():
            continue
        k, v = line.rsplit(" ", 1)
        out[k] = float(v)
    return out


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_metrics_tcp_endpoint(tmp_path):
    """The chart serves metrics over TCP (0.0.0.0:9400); cover that path."""
    import urllib.request
    root = build_tree(tmp_path / "sys", n_gpus=1)
    sock = str(tmp_path / "amd.sock")
    port = free_port()
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", sock, "--no-register",
         "--metrics-addr", f"127.0.0.1:{port}", "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 10
        body = None
        while time.time() < deadline and proc.poll() is None:
            try:
                body = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/metrics", timeout=5
                ).read().decode()
                break
            except OSError:
                time.sleep(0.1)
        assert body is not None, "metrics endpoint never came up"
        m = parse_metrics(body)
        assert m['k3samd_gpu_devices{health="healthy"}'] == 1
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_metrics_endpoint(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=2)
    sock = str(tmp_path / "amd.sock")
    msock = str(tmp_path / "metrics.sock")
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", sock, "--no-register",
         "--metrics-addr", f"unix:{msock}", "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 10
        while not (Path(sock).exists() and Path(msock).exists()):
            assert time.time() < deadline and proc.poll() is None
            time.sleep(0.05)
        _, body = scrape(msock)
        m = parse_metrics(body)
        assert m['k3samd_gpu_devices{health="healthy"}'] == 2
        assert m["k3samd_allocations_total"] == 0
        # per-GPU runtime gauges from the fixture's DRM tree
        gpu0 = 'gpu="amdgpu-1a2b3c4d5e6f0000"'
        assert m[f"k3samd_gpu_busy_percent{{{gpu0}}}"] == 0
        assert m[f"k3samd_gpu_temp_celsius{{{gpu0}}}"] == 53.0
        assert m[f"k3samd_gpu_power_watts{{{gpu0}}}"] == 135.0
        assert m[f"k3samd_gpu_vram_used_bytes{{{gpu0}}}"] == 2 * (1 << 20)

        # drive one allocation through grpc and re-scrape
        import grpc
        ch = grpc.insecure_channel(f"unix:{sock}")
        law = ch.unary_stream("/v1beta1.DevicePlugin/ListAndWatch",
                              request_serializer=IDENT,
                              response_deserializer=IDENT)(b"", timeout=10)
        devs = pb.decode_list_and_watch(next(law))
        alloc = ch.unary_unary("/v1beta1.DevicePlugin/Allocate",
                               request_serializer=IDENT,
                               response_deserializer=IDENT)
        alloc(pb.encode_allocate_request([[devs[0]["id"], devs[1]["id"]]]),
              timeout=10)
        try:
            alloc(pb.encode_allocate_request([["bogus"]]), timeout=10)
        except grpc.RpcError:
            pass
        ch.close()

        _, body = scrape(msock)
        m = parse_metrics(body)
        assert m["k3samd_allocations_total"] == 1
        assert m["k3samd_allocated_devices_total"] == 2
        assert m["k3samd_allocation_errors_total"] == 1
        assert m["k3samd_list_and_watch_updates_total"] >= 1
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_silent_client_does_not_wedge(tmp_path):
    """A client that connects and sends nothing must not block the
    livenessProbe: while the silent connection is still open, a normal
    scrape has to succeed promptly (each connection is served on its own
    thread with a 2 s socket timeout)."""
    import urllib.request
    root = build_tree(tmp_path / "sys", n_gpus=1)
    sock = str(tmp_path / "amd.sock")
    port = free_port()
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", sock, "--no-register",
         "--metrics-addr", f"127.0.0.1:{port}", "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    silent = []
    try:
        deadline = time.time() + 10
        while time.time() < deadline:
            try:
                s = socket.create_connection(("127.0.0.1", port), timeout=2)
                silent.append(s)  # connect, send nothing, keep it open
                break
            except OSError:
                time.sleep(0.1)
        assert silent, "endpoint never came up"
        # two more silent clients for good measure
        for _ in range(2):
            silent.append(
                socket.create_connection(("127.0.0.1", port), timeout=2))
        t0 = time.time()
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=5).read().decode()
        assert "k3samd_gpu_devices" in body
        assert time.time() - t0 < 3, "scrape was blocked by silent clients"
    finally:
        for s in silent:
            s.close()
        proc.terminate()
        proc.wait(timeout=10)


def test_ras_gauges_exposed(tmp_path):
    """The health model's RAS counters appear as per-GPU gauges so an
    operator can alert before a GPU flips Unhealthy."""
    import urllib.request
    from sysfs_builder import make_gpu_sick
    root = build_tree(tmp_path / "sys", n_gpus=1)
    make_gpu_sick(root, 0, umc_ue=2, umc_ce=41, resets=1)
    sock = str(tmp_path / "amd.sock")
    port = free_port()
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", sock, "--no-register",
         "--metrics-addr", f"127.0.0.1:{port}", "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 10
        body = None
        while time.time() < deadline and proc.poll() is None:
            try:
                body = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/metrics", timeout=5
                ).read().decode()
                break
            except OSError:
                time.sleep(0.1)
        assert body
        m = parse_metrics(body)
        gid = "amdgpu-1a2b3c4d5e6f0000"
        assert m[f'k3samd_gpu_ras_uncorrectable_errors{{gpu="{gid}"}}'] == 2
        assert m[f'k3samd_gpu_ras_correctable_errors{{gpu="{gid}"}}'] == 41
        assert m[f'k3samd_gpu_reset_count{{gpu="{gid}"}}'] == 1
    finally:
        proc.terminate()
        proc.wait(timeout=10)

"""Native unit selftest (HPACK RFC 7541 vectors, proto, yaml, json) and the
ThreadSanitizer race check over the hand-rolled gRPC threading."""

import subprocess
import time
from concurrent import futures
from pathlib import Path

import grpc
import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "native" / "bin"

IDENT = lambda b: b  # noqa: E731


def test_selftest_passes():
    proc = subprocess.run([str(BIN / "k3samd-selftest")],
                          capture_output=True, text=True, timeout=120)
    assert proc.returncode == 0, proc.stderr
    assert "all checks passed" in proc.stdout


def test_selftest_asan():
    """Memory-safety pass over the parsers (HPACK/yaml/json/proto)."""
    proc = subprocess.run(["make", "-C", str(REPO / "native"), "asan"],
                          capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stdout + proc.stderr
    assert "all checks passed" in proc.stdout


def test_tsan_grpc_exercise(tmp_path):
    """Drive the TSan build of the plugin through concurrent RPCs (options,
    allocate, a live ListAndWatch stream + health flips) and require zero
    ThreadSanitizer reports."""
    subprocess.run(["make", "-C", str(REPO / "native"), "tsan"], check=True,
                   capture_output=True)
    root = build_tree(tmp_path / "sys", n_gpus=4)
    sock = str(tmp_path / "amd.sock")
    proc = subprocess.Popen(
        [str(BIN / "k3samd-device-plugin-tsan"), "--plugin-sock", sock,
         "--no-register", "--health-poll-ms", "50"],
        env={"K3SAMD_SYSFS_ROOT": str(root),
             "TSAN_OPTIONS": "exitcode=66 halt_on_error=0"},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    try:
        deadline = time.time() + 15
        while not Path(sock).exists():
            if proc.poll() is not None:
                err = proc.stderr.read().decode(errors="replace")
                if "WARNING: ThreadSanitizer" not in err:
                    # gcc-11 libtsan cannot start on some kernels (ASLR
                    # "unexpected memory mapping") — environment, not a race
                    pytest.skip(f"libtsan unusable here: {err[:200]}")
                pytest.fail(f"tsan plugin died at startup: {err[:2000]}")
            assert time.time() < deadline
            time.sleep(0.05)
        channel = grpc.insecure_channel(f"unix:{sock}")
        law = channel.unary_stream("/v1beta1.DevicePlugin/ListAndWatch",
                                   request_serializer=IDENT,
                                   response_deserializer=IDENT)(b"", timeout=20)
        devs = pb.decode_list_and_watch(next(law))
        ids = [d["id"] for d in devs]

        def hammer(_):
            ch = grpc.insecure_channel(f"unix:{sock}")
            opt = ch.unary_unary("/v1beta1.DevicePlugin/GetDevicePluginOptions",
                                 request_serializer=IDENT,
                                 response_deserializer=IDENT)
            alloc = ch.unary_unary("/v1beta1.DevicePlugin/Allocate",
                                   request_serializer=IDENT,
                                   response_deserializer=IDENT)
            for _ in range(10):
                pb.decode_options(opt(b"", timeout=10))
                alloc(pb.encode_allocate_request([[ids[0]]]), timeout=10)
            ch.close()

        with futures.ThreadPoolExecutor(max_workers=4) as ex:
            list(ex.map(hammer, range(4)))
        channel.close()
    finally:
        proc.terminate()
        try:
            out, err = proc.communicate(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
            out, err = proc.communicate()
    text = err.decode(errors="replace")
    assert "WARNING: ThreadSanitizer" not in text, text[:4000]
    assert proc.returncode != 66, text[:4000]

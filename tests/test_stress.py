"""Longevity/robustness: connection churn must not leak fds or threads,
and time-sliced co-scheduling (4 pods on one GPU) must allocate cleanly."""

import os
import subprocess
import time
from pathlib import Path

import grpc
import pytest

import pb_v1beta1 as pb
from test_deviceplugin import PluginHarness

IDENT = lambda b: b  # noqa: E731

REPO = Path(__file__).resolve().parent.parent


def fd_count(pid):
    return len(os.listdir(f"/proc/{pid}/fd"))


def test_connection_churn_no_fd_leak(tmp_path):
    h = PluginHarness(tmp_path, n_gpus=2, replicas=1, register=False)
    try:
        def one_round():
            ch = grpc.insecure_channel(f"unix:{h.plugin_sock}")
            opt = ch.unary_unary(
                "/v1beta1.DevicePlugin/GetDevicePluginOptions",
                request_serializer=IDENT, response_deserializer=IDENT)
            pb.decode_options(opt(b"", timeout=10))
            ch.close()

        for _ in range(10):
            one_round()
        time.sleep(0.3)
        baseline = fd_count(h.proc.pid)
        for _ in range(50):
            one_round()
        # close-side threads lag under load: poll rather than fixed-sleep
        deadline = time.time() + 15
        while True:
            after = fd_count(h.proc.pid)
            if after <= baseline + 3 or time.time() > deadline:
                break
            time.sleep(0.2)
        assert after <= baseline + 3, (baseline, after)
    finally:
        h.close()


def test_timeslice_coscheduling_four_pods(tmp_path):
    """values.yaml:17-18 / README.md:112 behavior: four pods each get one
    replica of the same physical GPU; each Allocate succeeds and maps to
    the same physical device."""
    h = PluginHarness(tmp_path, n_gpus=1, replicas=4, register=False)
    try:
        devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
        assert len(devs) == 4
        minors = set()
        for d in devs:  # kubelet hands each pod a distinct replica id
            resp = pb.decode_allocate_response(
                h.call("Allocate", pb.encode_allocate_request([[d["id"]]])))
            minors.add(resp[0]["envs"]["K3SAMD_RENDER_MINORS"])
        assert minors == {"128"}  # all four land on the one physical GPU
    finally:
        h.close()


def test_many_listandwatch_streams(tmp_path):
    """kubelet reconnects create fresh streams; old ones must die cleanly."""
    h = PluginHarness(tmp_path, n_gpus=1, replicas=1, register=False)
    try:
        for _ in range(10):
            stream = h.stream("ListAndWatch")
            devs = pb.decode_list_and_watch(next(stream))
            assert len(devs) == 1
            stream.cancel()
        time.sleep(0.5)
        # plugin still healthy
        opts = pb.decode_options(h.call("GetDevicePluginOptions"))
        assert opts["get_preferred_allocation_available"]
    finally:
        h.close()


def test_connection_flood_bounded(tmp_path):
    """A flood of idle connections must not grow plugin threads without
    bound: excess connections beyond the cap are dropped, and the
    (single) legitimate kubelet connection keeps working."""
    import socket as socket_mod
    from test_deviceplugin import PluginHarness
    import pb_v1beta1 as pb
    h = PluginHarness(tmp_path, n_gpus=1, replicas=1, register=False)
    # grpcio connects lazily: make the legitimate connection FIRST so it
    # owns a slot before the flood arrives (kubelet's long-lived
    # connection predates any attack in practice)
    opts = pb.decode_options(h.call("GetDevicePluginOptions", timeout=10))
    assert opts["get_preferred_allocation_available"] is True
    floods = []
    try:
        for _ in range(60):
            try:
                s = socket_mod.socket(socket_mod.AF_UNIX,
                                      socket_mod.SOCK_STREAM)
                s.connect(h.plugin_sock)
                floods.append(s)
            except OSError:
                break
        time.sleep(0.3)
        # the established client still gets served mid-flood
        opts = pb.decode_options(h.call("GetDevicePluginOptions", timeout=10))
        assert opts["get_preferred_allocation_available"] is True
        # thread count is bounded: /proc/<pid>/task
        tasks = len(list(Path(f"/proc/{h.proc.pid}/task").iterdir()))
        assert tasks < 64, f"{tasks} threads after flood"
    finally:
        for s in floods:
            s.close()
        h.close()

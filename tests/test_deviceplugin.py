"""gRPC interop tests: the C++ device plugin against grpcio (the same gRPC
core family kubelet's grpc-go belongs to).

Covers the reference stack's scheduling surface end-to-end without a
cluster: registration handshake, ListAndWatch device
advertisement with time-slice replica fan-out (values.yaml:12-18 semantics),
Allocate device injection, preferred allocation, failRequestsGreaterThanOne,
and health transitions pushed over the live ListAndWatch stream.
"""

import json
import shutil
import subprocess
import threading
import time
from concurrent import futures
from pathlib import Path

import grpc
import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"

IDENT = lambda b: b  # noqa: E731  identity (de)serializer


class FakeKubelet:
    """Records Register calls, like kubelet's Registration service."""

    def __init__(self, sock_path):
        # like the real kubelet, claim the socket path on startup
        Path(sock_path).unlink(missing_ok=True)
        self.requests = []
        self.event = threading.Event()
        handler = grpc.method_handlers_generic_handler(
            "v1beta1.Registration",
            {"Register": grpc.unary_unary_rpc_method_handler(
                self._register,
                request_deserializer=IDENT, response_serializer=IDENT)})
        self.server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        self.server.add_generic_rpc_handlers((handler,))
        self.server.add_insecure_port(f"unix:{sock_path}")
        self.server.start()

    def _register(self, request, context):
        self.requests.append(pb.decode_register_request(request))
        self.event.set()
        return b""

    def stop(self):
        self.server.stop(0)


DEFAULT_CFG = """\
version: v1
flags:
  migStrategy: none
sharing:
  timeSlicing:
    renameByDefault: false
    failRequestsGreaterThanOne: false
    resources:
    - name: amd.com/gpu
      replicas: {replicas}
"""


class PluginHarness:
    def __init__(self, tmp_path, n_gpus=8, replicas=1, cfg_text=None,
                 register=True, health_poll_ms=200):
        self.root = build_tree(tmp_path / "sys", n_gpus=n_gpus)
        cfg = tmp_path / "config.yaml"
        cfg.write_text(cfg_text or DEFAULT_CFG.format(replicas=replicas))
        self.plugin_sock = str(tmp_path / "amd-gpu.sock")
        self.kubelet_sock = str(tmp_path / "kubelet.sock")
        self.kubelet = FakeKubelet(self.kubelet_sock) if register else None
        argv = [str(PLUGIN), "--config", str(cfg),
                "--plugin-sock", self.plugin_sock,
                "--health-poll-ms", str(health_poll_ms)]
        if register:
            argv += ["--kubelet-sock", self.kubelet_sock]
        else:
            argv += ["--no-register"]
        self.proc = subprocess.Popen(
            argv, env={"K3SAMD_SYSFS_ROOT": str(self.root)},
            stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        deadline = time.time() + 10
        while not Path(self.plugin_sock).exists():
            if time.time() > deadline or self.proc.poll() is not None:
                raise RuntimeError(
                    f"plugin did not start: {self.proc.stderr.read()}")
            time.sleep(0.02)
        self.channel = grpc.insecure_channel(f"unix:{self.plugin_sock}")

    def call(self, method, request=b"", timeout=5):
        fn = self.channel.unary_unary(f"/v1beta1.DevicePlugin/{method}",
                                      request_serializer=IDENT,
                                      response_deserializer=IDENT)
        return fn(request, timeout=timeout)

    def stream(self, method, request=b"", timeout=10):
        fn = self.channel.unary_stream(f"/v1beta1.DevicePlugin/{method}",
                                       request_serializer=IDENT,
                                       response_deserializer=IDENT)
        return fn(request, timeout=timeout)

    def close(self):
        self.channel.close()
        self.proc.terminate()
        try:
            self.proc.wait(timeout=5)
        except subprocess.TimeoutExpired:
            self.proc.kill()
        if self.kubelet:
            self.kubelet.stop()


@pytest.fixture
def harness(tmp_path):
    hs = []

    def make(**kw):
        h = PluginHarness(tmp_path, **kw)
        hs.append(h)
        return h

    yield make
    for h in hs:
        h.close()


def test_register_handshake(harness):
    h = harness(n_gpus=8, replicas=1)
    assert h.kubelet.event.wait(5), "plugin never registered"
    req = h.kubelet.requests[0]
    assert req["version"] == "v1beta1"
    assert req["endpoint"] == "amd-gpu.sock"
    assert req["resource_name"] == "amd.com/gpu"
    assert req["options"]["get_preferred_allocation_available"] is True


def test_prestart_container(harness):
    h = harness(n_gpus=1, register=False)
    assert h.call("PreStartContainer", b"") == b""


def test_options(harness):
    h = harness(n_gpus=2, register=False)
    opts = pb.decode_options(h.call("GetDevicePluginOptions"))
    assert opts["get_preferred_allocation_available"] is True
    assert opts["pre_start_required"] is False


def test_list_and_watch_exclusive(harness):
    h = harness(n_gpus=8, replicas=1, register=False)
    stream = h.stream("ListAndWatch")
    devs = pb.decode_list_and_watch(next(stream))
    assert len(devs) == 8
    assert all(d["health"] == "Healthy" for d in devs)
    assert all(d["id"].startswith("amdgpu-") for d in devs)
    assert {d["numa"] for d in devs} == {0, 1}  # fixture alternates sockets


def test_list_and_watch_timesliced(harness):
    """values.yaml:17-18 semantics: replicas=4 => 4 virtual devices per GPU
    (32 allocatable on the 8-GPU node, BASELINE.md structural target)."""
    h = harness(n_gpus=8, replicas=4, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    assert len(devs) == 32
    suffixes = [d["id"].rsplit("::", 1)[1] for d in devs]
    assert suffixes.count("0") == 8 and suffixes.count("3") == 8


def test_allocate_single(harness):
    h = harness(n_gpus=8, replicas=1, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    resp = pb.decode_allocate_response(
        h.call("Allocate", pb.encode_allocate_request([[devs[0]["id"]]])))
    assert len(resp) == 1
    cr = resp[0]
    paths = {d["host_path"] for d in cr["devices"]}
    assert "/dev/kfd" in paths
    assert "/dev/dri/renderD128" in paths
    assert "/dev/dri/card0" in paths
    assert cr["envs"]["K3SAMD_VISIBLE_DEVICES"] == devs[0]["id"]
    assert cr["envs"]["K3SAMD_RENDER_MINORS"] == "128"
    assert all(d["permissions"] == "rw" for d in cr["devices"])


def test_allocate_all_eight(harness):
    h = harness(n_gpus=8, replicas=1, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    ids = [d["id"] for d in devs]
    resp = pb.decode_allocate_response(
        h.call("Allocate", pb.encode_allocate_request([ids])))
    cr = resp[0]
    assert len(cr["devices"]) == 1 + 8 + 8  # kfd + 8 render + 8 card
    minors = cr["envs"]["K3SAMD_RENDER_MINORS"].split(",")
    assert sorted(minors) == sorted(str(m) for m in range(128, 136))


def test_allocate_replicas_dedupe(harness):
    """Two replicas of the same physical GPU inject that GPU once."""
    h = harness(n_gpus=2, replicas=4, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    base = devs[0]["id"].rsplit("::", 1)[0]
    ids = [f"{base}::0", f"{base}::1"]
    resp = pb.decode_allocate_response(
        h.call("Allocate", pb.encode_allocate_request([ids])))
    cr = resp[0]
    render = [d for d in cr["devices"] if "renderD" in d["host_path"]]
    assert len(render) == 1


def test_allocate_unknown_id(harness):
    h = harness(n_gpus=1, register=False)
    with pytest.raises(grpc.RpcError) as ei:
        h.call("Allocate", pb.encode_allocate_request([["nope"]]))
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT


def test_fail_requests_greater_than_one(harness):
    cfg = DEFAULT_CFG.format(replicas=4).replace(
        "failRequestsGreaterThanOne: false",
        "failRequestsGreaterThanOne: true")
    h = harness(n_gpus=2, cfg_text=cfg, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    ids = [devs[0]["id"], devs[1]["id"]]
    with pytest.raises(grpc.RpcError) as ei:
        h.call("Allocate", pb.encode_allocate_request([ids]))
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    # single-device requests still succeed
    resp = pb.decode_allocate_response(
        h.call("Allocate", pb.encode_allocate_request([[devs[0]["id"]]])))
    assert resp[0]["devices"]


def test_rename_by_default(harness):
    cfg = DEFAULT_CFG.format(replicas=4).replace(
        "renameByDefault: false", "renameByDefault: true")
    h = harness(n_gpus=1, cfg_text=cfg)
    assert h.kubelet.event.wait(5)
    assert h.kubelet.requests[0]["resource_name"] == "amd.com/gpu.shared"


def test_allocate_multi_container_pod(harness):
    """A pod with two GPU containers sends one AllocateRequest with two
    container_requests; each gets its own response."""
    h = harness(n_gpus=4, replicas=1, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    req = pb.encode_allocate_request([[devs[0]["id"]], [devs[1]["id"]]])
    resp = pb.decode_allocate_response(h.call("Allocate", req))
    assert len(resp) == 2
    assert resp[0]["envs"]["K3SAMD_RENDER_MINORS"] == "128"
    assert resp[1]["envs"]["K3SAMD_RENDER_MINORS"] == "129"


def test_preferred_allocation_must_include(harness):
    h = harness(n_gpus=2, replicas=4, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    avail = [d["id"] for d in devs]
    must = [avail[5]]  # a replica of GPU 1
    resp = pb.decode_preferred_response(
        h.call("GetPreferredAllocation",
               pb.encode_preferred_request(avail, must, 2)))
    assert len(resp[0]) == 2
    assert must[0] in resp[0]


def test_preferred_allocation_packs_replicas(harness):
    h = harness(n_gpus=2, replicas=4, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    avail = [d["id"] for d in devs]
    resp = pb.decode_preferred_response(
        h.call("GetPreferredAllocation",
               pb.encode_preferred_request(avail, [], 3)))
    assert len(resp) == 1 and len(resp[0]) == 3
    # all three picks should share one physical GPU
    bases = {i.rsplit("::", 1)[0] for i in resp[0]}
    assert len(bases) == 1


def test_preferred_allocation_numa_affine_exclusive(harness):
    """Exclusive multi-GPU requests prefer GPUs on the must-include set's
    host NUMA node (fixture: even GPU indices on node 0, odd on node 1;
    xGMI is uniform, host-side locality is what differs)."""
    h = harness(n_gpus=8, replicas=1, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    avail = [d["id"] for d in devs]
    must = [avail[1]]  # GPU 1 -> NUMA 1
    resp = pb.decode_preferred_response(
        h.call("GetPreferredAllocation",
               pb.encode_preferred_request(avail, must, 4)))
    chosen = resp[0]
    assert len(chosen) == 4 and must[0] in chosen
    # all four picks are odd-indexed GPUs (NUMA node 1)
    idx = [int(c[-1], 16) for c in chosen]
    assert all(i % 2 == 1 for i in idx), chosen


def test_preferred_allocation_numa_majority_without_must(harness):
    """With no must-include, pick the NUMA node that can satisfy the
    request with the most locality: 3 free GPUs on node 0 vs 1 on node 1
    -> a 3-GPU request lands entirely on node 0."""
    h = harness(n_gpus=8, replicas=1, register=False)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    ids = [d["id"] for d in devs]
    avail = [ids[0], ids[2], ids[4], ids[1]]  # three numa-0, one numa-1
    resp = pb.decode_preferred_response(
        h.call("GetPreferredAllocation",
               pb.encode_preferred_request(avail, [], 3)))
    idx = [int(c[-1], 16) for c in resp[0]]
    assert sorted(idx) == [0, 2, 4], resp[0]


def test_health_transition_pushes_update(harness):
    h = harness(n_gpus=2, replicas=1, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    first = pb.decode_list_and_watch(next(stream))
    assert all(d["health"] == "Healthy" for d in first)
    # GPU 1 dies: its KFD node vanishes from sysfs
    shutil.rmtree(h.root / "class/kfd/kfd/topology/nodes/3")
    second = pb.decode_list_and_watch(next(stream))
    by_health = sorted(d["health"] for d in second)
    assert by_health == ["Healthy", "Unhealthy"]


def test_sick_gpu_ras_ue_marked_unhealthy(harness):
    """Sick-but-PRESENT GPU: an uncorrectable
    RAS error (HBM ECC UE) must flip the device to Unhealthy over the
    live ListAndWatch even though its KFD node still exists — kubelet
    then stops scheduling pods onto it."""
    from sysfs_builder import make_gpu_sick
    h = harness(n_gpus=2, replicas=1, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    first = pb.decode_list_and_watch(next(stream))
    assert all(d["health"] == "Healthy" for d in first)
    make_gpu_sick(h.root, card_index=1, umc_ue=1)
    second = pb.decode_list_and_watch(next(stream))
    health = {d["id"]: d["health"] for d in second}
    assert sorted(health.values()) == ["Healthy", "Unhealthy"]
    # the SICK gpu (card 1 = unique_id ...0001) is the unhealthy one
    sick_id = [i for i, hl in health.items() if hl == "Unhealthy"][0]
    assert sick_id.endswith("0001")
    # healing (RAS counters reset, e.g. after page retirement + reboot)
    make_gpu_sick(h.root, card_index=1, umc_ue=0)
    third = pb.decode_list_and_watch(next(stream))
    assert all(d["health"] == "Healthy" for d in third)


def test_sick_gpu_reset_event_marked_unhealthy(harness):
    """A completed GPU reset (recovered hang) should drain the device."""
    from sysfs_builder import make_gpu_sick
    h = harness(n_gpus=1, replicas=2, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    first = pb.decode_list_and_watch(next(stream))
    assert len(first) == 2
    make_gpu_sick(h.root, card_index=0, resets=1)
    second = pb.decode_list_and_watch(next(stream))
    # BOTH time-slice replicas of the reset GPU go Unhealthy
    assert [d["health"] for d in second] == ["Unhealthy", "Unhealthy"]


def test_sick_gpu_aca_format_marked_unhealthy(harness):
    """The MI355X's actual RAS sysfs generation (ACA banks: ras/aca_umc
    with ue:/ce:/de: lines — format captured from real silicon in
    gpurun_out/r2_aca_probe.txt). A deferred (poison-pending) error must
    flip the device Unhealthy under the default policy."""
    from sysfs_builder import make_gpu_sick
    h = harness(n_gpus=2, replicas=1, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    assert all(d["health"] == "Healthy"
               for d in pb.decode_list_and_watch(next(stream)))
    make_gpu_sick(h.root, card_index=0, aca=True, umc_de=1)
    second = pb.decode_list_and_watch(next(stream))
    health = {d["id"]: d["health"] for d in second}
    assert sorted(health.values()) == ["Healthy", "Unhealthy"]
    sick = [i for i, hl in health.items() if hl == "Unhealthy"][0]
    assert sick.endswith("0000")


def test_fatal_ras_event_marked_unhealthy(harness):
    """ras/event_state 'Fatal Error: count:N' (the MI355X reset/poison
    channel) drains the GPU."""
    from sysfs_builder import make_gpu_sick
    h = harness(n_gpus=1, replicas=1, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    assert pb.decode_list_and_watch(next(stream))[0]["health"] == "Healthy"
    make_gpu_sick(h.root, card_index=0, aca=True, fatal_events=1)
    assert pb.decode_list_and_watch(next(stream))[0]["health"] == "Unhealthy"


def test_health_thresholds_configurable(harness):
    """Correctable-error budget comes from the additive `health:` config
    block; under the threshold stays Healthy, over it flips."""
    from sysfs_builder import make_gpu_sick
    cfg = DEFAULT_CFG.format(replicas=1) + (
        "health:\n"
        "  maxCorrectableErrors: 100\n"
        "  maxResets: -1\n")
    h = harness(n_gpus=1, replicas=1, register=False, health_poll_ms=100,
                cfg_text=cfg)
    stream = h.stream("ListAndWatch", timeout=30)
    assert pb.decode_list_and_watch(next(stream))[0]["health"] == "Healthy"
    make_gpu_sick(h.root, card_index=0, umc_ce=50, gfx_ce=30)  # 80 <= 100
    time.sleep(0.5)  # two poll periods; no update should be pushed
    make_gpu_sick(h.root, card_index=0, umc_ce=90, gfx_ce=30)  # 120 > 100
    second = pb.decode_list_and_watch(next(stream))
    assert second[0]["health"] == "Unhealthy"


def test_config_hot_reload(harness, tmp_path):
    """Editing the mounted config (kubelet updates ConfigMap mounts in
    place) re-fans-out the device list and pushes it over the live
    ListAndWatch stream — allocatable changes without a pod restart."""
    h = harness(n_gpus=2, replicas=1, register=False, health_poll_ms=100)
    stream = h.stream("ListAndWatch", timeout=30)
    first = pb.decode_list_and_watch(next(stream))
    assert len(first) == 2
    cfg = tmp_path / "config.yaml"
    time.sleep(0.05)
    cfg.write_text(DEFAULT_CFG.format(replicas=4))
    second = pb.decode_list_and_watch(next(stream))
    assert len(second) == 8
    assert all("::" in d["id"] for d in second)
    # invalid update is ignored, plugin stays up with the last-good config
    time.sleep(0.05)
    cfg.write_text("version: v99\n")
    time.sleep(0.5)
    devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
    assert len(devs) == 8


def test_hot_reload_rename_reregisters(harness, tmp_path):
    '''Turning renameByDefault on via config reload changes the advertised
    resource, which requires a fresh Register with kubelet.'''
    h = harness(n_gpus=1, replicas=4, register=True, health_poll_ms=100)
    assert h.kubelet.event.wait(5)
    assert h.kubelet.requests[0]["resource_name"] == "amd.com/gpu"
    h.kubelet.event.clear()
    cfg = tmp_path / "config.yaml"
    cfg.write_text(DEFAULT_CFG.format(replicas=4).replace(
        "renameByDefault: false", "renameByDefault: true"))
    assert h.kubelet.event.wait(10), "no re-register after rename"
    assert h.kubelet.requests[-1]["resource_name"] == "amd.com/gpu.shared"


def test_reregister_after_kubelet_restart(harness):
    """kubelet restarts drop plugin registrations; the plugin must notice
    the new registration socket and Register again."""
    h = harness(n_gpus=1, replicas=1)
    assert h.kubelet.event.wait(5)
    assert len(h.kubelet.requests) == 1
    # simulate kubelet restart: new server process on the same socket path
    h.kubelet.stop()
    time.sleep(0.2)
    h.kubelet = FakeKubelet(h.kubelet_sock)
    assert h.kubelet.event.wait(10), "plugin did not re-register"
    assert h.kubelet.requests[0]["resource_name"] == "amd.com/gpu"


def test_socket_takeover_on_restart(tmp_path):
    '''A replacement plugin instance (DaemonSet restart with a stale pod
    lingering) takes over the socket path; clients reach the new one.'''
    root = build_tree(tmp_path / "sys", n_gpus=1)
    sock = str(tmp_path / "amd.sock")
    env = {"K3SAMD_SYSFS_ROOT": str(root)}
    p1 = subprocess.Popen([str(PLUGIN), "--plugin-sock", sock,
                           "--no-register"], env=env,
                          stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    p2 = None
    try:
        deadline = time.time() + 10
        while not Path(sock).exists():
            assert time.time() < deadline and p1.poll() is None
            time.sleep(0.02)
        p2 = subprocess.Popen([str(PLUGIN), "--plugin-sock", sock,
                               "--no-register"], env=env,
                              stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        time.sleep(0.5)
        assert p2.poll() is None, p2.stderr.read()[:300]
        ch = grpc.insecure_channel(f"unix:{sock}")
        opts = pb.decode_options(
            ch.unary_unary("/v1beta1.DevicePlugin/GetDevicePluginOptions",
                           request_serializer=IDENT,
                           response_deserializer=IDENT)(b"", timeout=10))
        assert opts["get_preferred_allocation_available"] is True
        ch.close()
    finally:
        for p in (p1, p2):
            if p is not None:
                p.terminate()
                p.wait(timeout=10)


def test_register_failure_exits_nonzero(tmp_path):
    '''No kubelet at the socket => serve() fails and the daemon exits 1
    (DaemonSet backoff semantics, not a silent half-start).'''
    root = build_tree(tmp_path / "sys", n_gpus=1)
    proc = subprocess.run(
        [str(PLUGIN), "--plugin-sock", str(tmp_path / "amd.sock"),
         "--kubelet-sock", str(tmp_path / "missing-kubelet.sock")],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        capture_output=True, text=True, timeout=60)
    assert proc.returncode == 1
    assert "Register with kubelet failed" in proc.stderr
    assert not (tmp_path / "amd.sock").exists()  # socket cleaned up


def test_cpu_only_zero_allocatable(tmp_path):
    """BASELINE.json config #1: CPU-only node => 0 amd.com/gpu."""
    root = build_tree(tmp_path / "sys", n_gpus=0)
    out = subprocess.run(
        [str(PLUGIN), "--oneshot"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        capture_output=True, text=True, timeout=60)
    j = json.loads(out.stdout)
    assert j["resource"] == "amd.com/gpu"
    assert j["allocatable"] == 0


def test_bad_config_rejected(tmp_path):
    cfg = tmp_path / "bad.yaml"
    cfg.write_text("version: v2\n")
    out = subprocess.run([str(PLUGIN), "--config", str(cfg), "--oneshot"],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 2
    assert "version" in out.stderr


def test_register_retries_until_kubelet_appears(tmp_path):
    """Node boot ordering: the plugin DaemonSet may start before kubelet's
    device-plugin registry socket exists. The plugin must retry with
    backoff and register once the (fake) kubelet comes up."""
    root = build_tree(tmp_path / "sys", n_gpus=1)
    kubelet_sock = str(tmp_path / "kubelet.sock")
    proc = subprocess.Popen(
        [str(PLUGIN), "--plugin-sock", str(tmp_path / "amd.sock"),
         "--kubelet-sock", kubelet_sock,
         "--register-retries", "20", "--register-backoff-ms", "200",
         "--health-poll-ms", "0"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    kubelet = None
    try:
        time.sleep(0.8)  # a few failed attempts happen first
        assert proc.poll() is None, proc.stderr.read().decode()
        kubelet = FakeKubelet(kubelet_sock)
        assert kubelet.event.wait(10), "plugin never registered"
        assert kubelet.requests[0]["resource_name"] == "amd.com/gpu"
        time.sleep(0.2)
        assert proc.poll() is None  # serving normally
    finally:
        if kubelet:
            kubelet.stop()
        proc.terminate()
        proc.wait(timeout=10)


def test_register_gives_up_after_retries(tmp_path):
    """Exhausted retries still end in a loud exit (DaemonSet backoff)."""
    root = build_tree(tmp_path / "sys", n_gpus=1)
    proc = subprocess.run(
        [str(PLUGIN), "--plugin-sock", str(tmp_path / "amd.sock"),
         "--kubelet-sock", str(tmp_path / "missing.sock"),
         "--register-retries", "2", "--register-backoff-ms", "50"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        capture_output=True, text=True, timeout=60)
    assert proc.returncode == 1
    assert "Register retry 2/2" in proc.stderr


def test_health_threshold_hot_reload(harness, tmp_path):
    """Tightening the health: thresholds via ConfigMap hot-reload drains
    an already-erroring GPU without a plugin restart."""
    from sysfs_builder import make_gpu_sick
    lenient = DEFAULT_CFG.format(replicas=1) + (
        "health:\n  maxCorrectableErrors: 1000\n  maxResets: -1\n")
    h = harness(n_gpus=1, replicas=1, register=False, health_poll_ms=100,
                cfg_text=lenient)
    stream = h.stream("ListAndWatch", timeout=30)
    assert pb.decode_list_and_watch(next(stream))[0]["health"] == "Healthy"
    make_gpu_sick(h.root, card_index=0, umc_ce=500)  # under 1000: healthy
    time.sleep(0.4)
    cfg = tmp_path / "config.yaml"
    time.sleep(0.05)
    cfg.write_text(DEFAULT_CFG.format(replicas=1) + (
        "health:\n  maxCorrectableErrors: 100\n  maxResets: -1\n"))
    # reload pushes a device-list update; the sick GPU must now be drained
    deadline = time.time() + 10
    while time.time() < deadline:
        devs = pb.decode_list_and_watch(next(stream))
        if devs[0]["health"] == "Unhealthy":
            break
    assert devs[0]["health"] == "Unhealthy"

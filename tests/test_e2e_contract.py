"""End-to-end Allocate -> OCI-injection contract test.

SURVEY.md §7 "hard parts" #1: the device-plugin-to-runtime handshake is
where bugs live. This test walks a pod's whole GPU path without a cluster:

  fake kubelet Register -> ListAndWatch -> (scheduler picks ids)
  -> Allocate -> returned env vars pasted into the pod's OCI config
  -> k3samd-oci-runtime transform -> the container sees exactly the
     allocated GPUs' device nodes, nothing more.
"""

import json
import subprocess
from pathlib import Path

import pytest

import pb_v1beta1 as pb
from test_deviceplugin import PluginHarness

REPO = Path(__file__).resolve().parent.parent
RUNTIME = REPO / "native" / "bin" / "k3samd-oci-runtime"


def oci_transform(tmp_path, sysfs_root, envs):
    spec = {
        "ociVersion": "1.0.2",
        "process": {"args": ["mi-stream"], "cwd": "/",
                    "env": [f"{k}={v}" for k, v in envs.items()]},
        "root": {"path": "rootfs"},
        "linux": {"resources": {"devices": [{"allow": False,
                                             "access": "rwm"}]}},
    }
    cfg = tmp_path / "config.json"
    cfg.write_text(json.dumps(spec))
    subprocess.run(
        [str(RUNTIME), "--transform-only", str(cfg)],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs_root),
             "K3SAMD_DEV_ROOT": str(tmp_path / "nodev")},
        check=True, capture_output=True, timeout=60)
    return json.loads(cfg.read_text())


def test_allocate_to_injection_single_gpu(tmp_path):
    h = PluginHarness(tmp_path, n_gpus=8, replicas=1, register=False)
    try:
        devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
        # "scheduler" picks GPU 5
        pick = devs[5]["id"]
        resp = pb.decode_allocate_response(
            h.call("Allocate", pb.encode_allocate_request([[pick]])))
        envs = resp[0]["envs"]
        out = oci_transform(tmp_path, h.root, envs)
        paths = [d["path"] for d in out["linux"]["devices"]]
        # exactly the allocated GPU: kfd + renderD133 + card5
        assert sorted(paths) == sorted(
            ["/dev/kfd", "/dev/dri/renderD133", "/dev/dri/card5"])
    finally:
        h.close()


def test_allocate_to_injection_timesliced_pair(tmp_path):
    h = PluginHarness(tmp_path, n_gpus=4, replicas=4, register=False)
    try:
        devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
        base0 = devs[0]["id"].rsplit("::", 1)[0]
        base1 = devs[4]["id"].rsplit("::", 1)[0]
        ids = [f"{base0}::2", f"{base1}::0"]  # replicas of two phys GPUs
        resp = pb.decode_allocate_response(
            h.call("Allocate", pb.encode_allocate_request([ids])))
        out = oci_transform(tmp_path, h.root, resp[0]["envs"])
        paths = sorted(d["path"] for d in out["linux"]["devices"])
        assert paths == sorted(["/dev/kfd",
                                "/dev/dri/renderD128", "/dev/dri/card0",
                                "/dev/dri/renderD129", "/dev/dri/card1"])
    finally:
        h.close()


def test_no_gpu_limit_pod_gets_nothing(tmp_path):
    """The scheduling-bypass hole closed: a pod that sets
    runtimeClassName: amd but requests NO amd.com/gpu limit never passes
    through Allocate, so its container has no allocation env/annotation —
    the runtime must inject zero GPU devices (default-deny), keeping
    kubelet's device accounting authoritative."""
    h = PluginHarness(tmp_path, n_gpus=8, replicas=1, register=False)
    try:
        # no Allocate call at all — this pod never requested the resource
        out = oci_transform(tmp_path, h.root, {"PATH": "/usr/bin"})
        assert "/dev/kfd" not in json.dumps(out)
        devs = out["linux"].get("devices", [])
        assert devs == []
        # the pre-existing default deny cgroup rule is untouched, and no
        # allow rules were added
        rules = out["linux"]["resources"]["devices"]
        assert rules == [{"allow": False, "access": "rwm"}]
    finally:
        h.close()

"""Control-plane binaries against the REAL sysfs of an MI355X box.

The CPU suite drives everything through fixture trees; these @gpu tests
prove the same binaries read real KFD topology correctly (the
"first run on real hardware is a validation, not a debug session" goal,
SURVEY.md §7)."""

import json
import subprocess
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "native" / "bin"

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="no GPU")


@requires_gpu
def test_device_plugin_oneshot_real_sysfs():
    out = subprocess.run([str(BIN / "k3samd-device-plugin"), "--oneshot"],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    j = json.loads(out.stdout)
    assert j["resource"] == "amd.com/gpu"
    assert j["allocatable"] >= 1
    assert all(d["healthy"] for d in j["devices"])


def real_major_minor(path):
    import os
    st = os.stat(path)
    return os.major(st.st_rdev), os.minor(st.st_rdev)


@requires_gpu
def test_cdi_gen_real_devices():
    out = subprocess.run([str(BIN / "k3samd-cdi-gen")],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    spec = json.loads(out.stdout)
    dev0 = next(d for d in spec["devices"] if d["name"] == "0")
    nodes = {n["path"]: n for n in dev0["containerEdits"]["deviceNodes"]}
    assert "/dev/kfd" in nodes
    render = [p for p in nodes if "renderD" in p]
    assert render
    # majors/minors must match the real device nodes (they are dynamic on
    # some kernels — this box reports e.g. 242, not the classic 226/10)
    for path in ["/dev/kfd", render[0]]:
        mj, mn = real_major_minor(path)
        assert (nodes[path]["major"], nodes[path]["minor"]) == (mj, mn), path


@requires_gpu
def test_labeller_real_sysfs():
    out = subprocess.run([str(BIN / "k3samd-node-labeller"), "--json"],
                         capture_output=True, text=True, timeout=60)
    labels = json.loads(out.stdout)
    assert labels["amd.com/gpu.present"] == "true"
    assert labels["amd.com/gpu.arch"] == "gfx950"
    assert labels["amd.com/gpu.family"] == "CDNA4"
    assert labels["amd.com/gpu.cu-count"] == "256"


@requires_gpu
def test_oci_transform_real_topology(tmp_path):
    spec = {
        "ociVersion": "1.0.2",
        "process": {"args": ["true"],
                    "env": ["K3SAMD_VISIBLE_DEVICES=all"], "cwd": "/"},
        "root": {"path": "rootfs"},
        "linux": {},
    }
    cfg = tmp_path / "config.json"
    cfg.write_text(json.dumps(spec))
    out = subprocess.run(
        [str(BIN / "k3samd-oci-runtime"), "--transform-only", str(cfg)],
        capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    j = json.loads(cfg.read_text())
    paths = [d["path"] for d in j["linux"]["devices"]]
    assert "/dev/kfd" in paths
    assert any("renderD" in p for p in paths)
    kfd = next(d for d in j["linux"]["devices"] if d["path"] == "/dev/kfd")
    mj, mn = real_major_minor("/dev/kfd")
    assert (kfd["major"], kfd["minor"]) == (mj, mn)

"""Node labeller tests (replaces NFD/GFD roles: README.md:97-103,
values.yaml:1-2; label surface mirrors nvidia.com/gpu.* incl. the
nodeSelector example nvidia-smi.yaml:6-7)."""

import json
import subprocess
from pathlib import Path

import pytest

from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
LABELLER = REPO / "native" / "bin" / "k3samd-node-labeller"


def run_labeller(sysfs_root, *args):
    return subprocess.run(
        [str(LABELLER), *args],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs_root)},
        capture_output=True, text=True, timeout=60)


def test_labels_eight_gpu(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=8)
    out = run_labeller(root, "--json")
    labels = json.loads(out.stdout)
    assert labels["amd.com/gpu.present"] == "true"
    assert labels["amd.com/gpu.count"] == "8"
    assert labels["amd.com/gpu.arch"] == "gfx950"
    assert labels["amd.com/gpu.family"] == "CDNA4"
    assert labels["amd.com/gpu.product"] == "AMD-Instinct-MI355X"
    assert labels["amd.com/gpu.vram"] == "294912Mi"  # 288 GiB
    assert labels["amd.com/gpu.xgmi-links"] == "7"
    assert labels["amd.com/gpu.cu-count"] == "256"
    assert labels["amd.com/gpu.driver-version"] == "6.12.12"


def test_features_file(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=2)
    ff = tmp_path / "features.d" / "k3samd"
    ff.parent.mkdir()
    out = run_labeller(root, "--oneshot", "--features-file", str(ff))
    assert out.returncode == 0
    lines = dict(l.split("=", 1) for l in ff.read_text().splitlines())
    assert lines["amd.com/gpu.present"] == "true"
    assert lines["amd.com/gpu.count"] == "2"
    assert not (tmp_path / "features.d" / "k3samd.tmp").exists()


def test_daemon_mode_writes_then_terminates(tmp_path):
    import signal
    import time
    root = build_tree(tmp_path / "sys", n_gpus=1)
    ff = tmp_path / "k3samd"
    proc = subprocess.Popen(
        [str(LABELLER), "--features-file", str(ff), "--interval-s", "1"],
        env={"K3SAMD_SYSFS_ROOT": str(root)})
    try:
        deadline = time.time() + 10
        while not ff.exists():
            assert time.time() < deadline and proc.poll() is None
            time.sleep(0.05)
        lines = dict(l.split("=", 1) for l in ff.read_text().splitlines())
        assert lines["amd.com/gpu.present"] == "true"
    finally:
        proc.send_signal(signal.SIGTERM)
        proc.wait(timeout=10)


def test_mixed_arch_uses_first_gpu(tmp_path):
    '''Heterogeneous nodes are out of scope for MI355X clusters, but the
    labeller must stay deterministic: labels describe the first GPU.'''
    root = build_tree(tmp_path / "sys", n_gpus=2)
    # mutate the second GPU's arch to gfx942
    props = root / "class/kfd/kfd/topology/nodes/3/properties"
    text = props.read_text().replace("gfx_target_version 90500",
                                     "gfx_target_version 90402")
    props.write_text(text)
    out = run_labeller(root, "--json")
    labels = json.loads(out.stdout)
    assert labels["amd.com/gpu.arch"] == "gfx950"   # first GPU
    assert labels["amd.com/gpu.count"] == "2"


def test_cpu_only_no_labels(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=0)
    ff = tmp_path / "k3samd"
    run_labeller(root, "--oneshot", "--features-file", str(ff))
    assert ff.read_text() == ""  # empty feature file => no gpu labels
    out = run_labeller(root, "--json")
    assert json.loads(out.stdout) == {}


def test_ras_capability_label(tmp_path):
    from sysfs_builder import make_gpu_sick
    root = build_tree(tmp_path / "sys", n_gpus=1)
    labels = json.loads(run_labeller(root, "--json").stdout)
    assert labels["amd.com/gpu.ras"] == "false"  # fixture has no RAS dir
    make_gpu_sick(root, 0, aca=True)  # creates ras/aca_* (all zeros)
    labels = json.loads(run_labeller(root, "--json").stdout)
    assert labels["amd.com/gpu.ras"] == "true"

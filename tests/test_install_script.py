"""install-node.sh behavior on a CPU-only box (no k3s): installs the
binaries to the prefix, warns about missing /dev/kfd, exits 0."""

import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def test_install_to_prefix(tmp_path):
    prefix = tmp_path / "bin"
    prefix.mkdir()
    proc = subprocess.run(
        ["sh", str(REPO / "deploy/scripts/install-node.sh"),
         "--no-build", f"--prefix={prefix}"],
        cwd=str(REPO), capture_output=True, text=True, timeout=120)
    assert proc.returncode == 0, proc.stderr
    installed = {p.name for p in prefix.iterdir()}
    assert {"k3samd-oci-runtime", "k3samd-device-plugin",
            "k3samd-node-labeller", "k3samd-cdi-gen", "mi355x-smi",
            "mi-stream", "mi-allreduce"} <= installed
    if not Path("/dev/kfd").exists():
        assert "WARNING" in proc.stdout
    if not Path("/var/lib/rancher/k3s").exists():
        assert "k3s not detected" in proc.stdout


def test_bad_flag_rejected():
    proc = subprocess.run(
        ["sh", str(REPO / "deploy/scripts/install-node.sh"), "--bogus"],
        cwd=str(REPO), capture_output=True, text=True, timeout=60)
    assert proc.returncode == 2

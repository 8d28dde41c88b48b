"""deploy/scripts/node-doctor.sh against synthetic node states."""

import os
import stat
import socket
import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
SCRIPT = REPO / "deploy" / "scripts" / "node-doctor.sh"


def run_doctor(tmp_path, *, kfd=True, renders=2, containerd=True,
               kubelet=True, plugin_sock=False):
    tmp_path = Path(tmp_path)
    tmp_path.mkdir(parents=True, exist_ok=True)
    dev = tmp_path / "dev"
    (dev / "dri").mkdir(parents=True)
    if kfd:
        (dev / "kfd").write_text("")
    for k in range(renders):
        (dev / "dri" / f"renderD{128 + k}").write_text("")
    cfg = tmp_path / "config.toml"
    if containerd:
        cfg.write_text(
            '[plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd]\n'
            '  runtime_type = "io.containerd.runc.v2"\n'
            '  [plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd.options]\n'
            '    BinaryName = "/usr/local/bin/k3samd-oci-runtime"\n')
    kdir = tmp_path / "kubelet"
    socks = []
    if kubelet:
        kdir.mkdir()
        s = socket.socket(socket.AF_UNIX)
        s.bind(str(kdir / "kubelet.sock"))
        socks.append(s)
        if plugin_sock:
            s2 = socket.socket(socket.AF_UNIX)
            s2.bind(str(kdir / "amd-gpu.sock"))
            socks.append(s2)
    # a fake installed runtime binary on PATH
    bindir = tmp_path / "bin"
    bindir.mkdir()
    rt = bindir / "k3samd-oci-runtime"
    rt.write_text("#!/bin/sh\nexit 0\n")
    rt.chmod(rt.stat().st_mode | stat.S_IEXEC)
    # fake mi355x-smi emitting the real (space-free) JSON shape — the
    # real box caught a sed that assumed a space after the colon
    smi = bindir / "mi355x-smi"
    smi.write_text('#!/bin/sh\necho \'{"driver_version":"x","gpu_count":2,"gpus":[]}\'\n')
    smi.chmod(smi.stat().st_mode | stat.S_IEXEC)
    env = dict(os.environ)
    env.update({
        "PATH": f"{bindir}:{env['PATH']}",
        "K3SAMD_DEV_ROOT": str(dev),
        "K3SAMD_CONTAINERD_CONFIG": str(cfg),
        "K3SAMD_KUBELET_DIR": str(kdir),
    })
    proc = subprocess.run(["sh", str(SCRIPT)], env=env, cwd=str(tmp_path),
                          capture_output=True, text=True, timeout=60)
    for s in socks:
        s.close()
    return proc


def test_ready_node_passes(tmp_path):
    proc = run_doctor(tmp_path / "n", plugin_sock=True)
    assert proc.returncode == 0, proc.stdout
    assert "node ready." in proc.stdout
    assert "FAIL" not in proc.stdout
    assert "enumerates 2 GPU(s)" in proc.stdout


def test_missing_kfd_fails(tmp_path):
    proc = run_doctor(tmp_path / "k", kfd=False)
    assert proc.returncode != 0
    assert "/dev/kfd missing" in proc.stdout


def test_unreadable_containerd_config_warns(tmp_path):
    proc = run_doctor(tmp_path / "a", containerd=False)
    # unreadable config is a warn (cannot verify), not a fail
    assert "cannot verify RuntimeClass" in proc.stdout


def test_wrong_runtime_entry_fails(tmp_path):
    proc = run_doctor(tmp_path / "a", plugin_sock=True)
    assert proc.returncode == 0
    d = tmp_path / "b"
    proc2 = run_doctor(d, plugin_sock=True)
    assert proc2.returncode == 0
    (d / "config.toml").write_text("[plugins]\n")
    import os as _os
    env = dict(_os.environ)
    env.update({
        "PATH": f"{d / 'bin'}:{env['PATH']}",
        "K3SAMD_DEV_ROOT": str(d / "dev"),
        "K3SAMD_CONTAINERD_CONFIG": str(d / "config.toml"),
        "K3SAMD_KUBELET_DIR": str(d / "kubelet"),
    })
    proc3 = subprocess.run(["sh", str(SCRIPT)], env=env, cwd=str(d),
                           capture_output=True, text=True, timeout=60)
    assert proc3.returncode != 0
    assert "lacks the 'amd' runtime entry" in proc3.stdout

"""Golden config.json tests for the OCI runtime wrapper.

Mirrors the reference's container-runtime layer contract
(/root/reference/README.md:57-69, :164): a pod with the GPU RuntimeClass
gets device nodes + cgroup rules (+ optionally ROCm userspace) injected,
scoped to the devices the device plugin allocated.
"""

import json
import subprocess
from pathlib import Path

import pytest

from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
RUNTIME = REPO / "native" / "bin" / "k3samd-oci-runtime"


BASE_SPEC = {
    "ociVersion": "1.0.2",
    "process": {
        "args": ["rocm-smi"],
        "env": ["PATH=/usr/bin"],
        "cwd": "/",
    },
    "root": {"path": "rootfs"},
    "mounts": [
        {"destination": "/proc", "type": "proc", "source": "proc"},
    ],
    "linux": {
        "namespaces": [{"type": "pid"}, {"type": "mount"}],
        "resources": {"devices": [{"allow": False, "access": "rwm"}]},
    },
}


def transform(tmp_path, spec, n_gpus=8, extra_env=None, runtime_env=None):
    sysfs = build_tree(tmp_path / "sys", n_gpus=n_gpus)
    cfg = tmp_path / "config.json"
    spec = json.loads(json.dumps(spec))  # deep copy
    if extra_env:
        spec["process"]["env"].extend(extra_env)
    cfg.write_text(json.dumps(spec))
    env = {"K3SAMD_SYSFS_ROOT": str(sysfs),
           "K3SAMD_DEV_ROOT": str(tmp_path / "nonexistent-dev")}
    if runtime_env:
        env.update(runtime_env)
    proc = subprocess.run(
        [str(RUNTIME), "--transform-only", str(cfg)],
        env=env, capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr
    return json.loads(cfg.read_text()), proc.stderr


def device_paths(spec):
    return [d["path"] for d in spec["linux"]["devices"]]


def test_no_allocation_injects_nothing_by_default(tmp_path):
    """Default-deny: runtimeClassName alone (no amd.com/gpu limit, so no
    Allocate env/annotation) must grant ZERO GPUs — otherwise any pod
    could bypass kubelet device accounting (the NVIDIA_VISIBLE_DEVICES=all
    foot-gun the reference stack inherits, /root/reference/README.md:164)."""
    out, err = transform(tmp_path, BASE_SPEC, n_gpus=8)
    assert "/dev/kfd" not in json.dumps(out)
    assert "devices" not in out["linux"] or not device_paths(out)
    assert "no allocation" in err and "default-deny" in err


def test_explicit_all_sentinel_injects_all_gpus(tmp_path):
    out, _ = transform(tmp_path, BASE_SPEC, n_gpus=8,
                       extra_env=["K3SAMD_VISIBLE_DEVICES=all"])
    paths = device_paths(out)
    assert "/dev/kfd" in paths
    for k in range(8):
        assert f"/dev/dri/renderD{128 + k}" in paths
        assert f"/dev/dri/card{k}" in paths
    assert len(paths) == 17
    # cgroup allow rules added alongside, preserving the existing deny rule
    rules = out["linux"]["resources"]["devices"]
    assert rules[0] == {"allow": False, "access": "rwm"}
    allows = [r for r in rules if r.get("allow")]
    assert len(allows) == 17
    for r in allows:
        assert r["type"] == "c" and r["access"] == "rwm"


def test_allow_all_runtime_flag(tmp_path):
    """Operator-level debug override: K3SAMD_ALLOW_ALL=1 on the runtime
    binary (NOT the container env) restores inject-all for no-allocation
    containers."""
    out, _ = transform(tmp_path, BASE_SPEC, n_gpus=2,
                       runtime_env={"K3SAMD_ALLOW_ALL": "1"})
    paths = device_paths(out)
    assert "/dev/kfd" in paths
    assert len(paths) == 5  # kfd + 2x(render+card)


def test_allow_all_not_honored_from_container_env(tmp_path):
    """K3SAMD_ALLOW_ALL inside the container's own env must NOT grant all
    GPUs — only the runtime binary's environment (operator-controlled)
    may; otherwise any pod spec could set it."""
    out, err = transform(tmp_path, BASE_SPEC, n_gpus=2,
                         extra_env=["K3SAMD_ALLOW_ALL=1"])
    assert "/dev/kfd" not in json.dumps(out)
    assert "no allocation" in err


def test_visible_devices_subset(tmp_path):
    env = ["K3SAMD_VISIBLE_DEVICES=amdgpu-1a2b3c4d5e6f0002",
           "K3SAMD_RENDER_MINORS=130"]
    out, _ = transform(tmp_path, BASE_SPEC, n_gpus=8, extra_env=env)
    paths = device_paths(out)
    assert paths == ["/dev/kfd", "/dev/dri/renderD130", "/dev/dri/card2"]


def test_visible_none_skips(tmp_path):
    out, err = transform(tmp_path, BASE_SPEC, n_gpus=8,
                         extra_env=["K3SAMD_VISIBLE_DEVICES=none"])
    assert "devices" not in out["linux"] or not any(
        "kfd" in p for p in device_paths(out))
    assert "skipped" in err


def test_unrelated_fields_roundtrip(tmp_path):
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"].append("K3SAMD_VISIBLE_DEVICES=all")
    spec["annotations"] = {"io.kubernetes.pod.name": "rocm-smi"}
    spec["hooks"] = {"prestart": [{"path": "/bin/true"}]}
    spec["linux"]["seccomp"] = {"defaultAction": "SCMP_ACT_ALLOW"}
    out, _ = transform(tmp_path, spec, n_gpus=1)
    assert out["annotations"] == spec["annotations"]
    assert out["hooks"] == spec["hooks"]
    assert out["linux"]["seccomp"] == spec["linux"]["seccomp"]
    assert out["mounts"][0]["destination"] == "/proc"
    assert out["process"]["args"] == ["rocm-smi"]


def test_idempotent(tmp_path):
    sysfs = build_tree(tmp_path / "sys", n_gpus=2)
    cfg = tmp_path / "config.json"
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"].append("K3SAMD_VISIBLE_DEVICES=all")
    cfg.write_text(json.dumps(spec))
    env = {"K3SAMD_SYSFS_ROOT": str(sysfs),
           "K3SAMD_DEV_ROOT": str(tmp_path / "nodev")}
    for _ in range(2):
        subprocess.run([str(RUNTIME), "--transform-only", str(cfg)],
                       env=env, check=True, capture_output=True, timeout=60)
    out = json.loads(cfg.read_text())
    paths = device_paths(out)
    assert len(paths) == len(set(paths)) == 5  # kfd + 2x(render+card)
    allows = [r for r in out["linux"]["resources"]["devices"]
              if r.get("allow")]
    assert len(allows) == 5


def test_rocm_mount_injection(tmp_path):
    rocm = tmp_path / "opt-rocm"
    rocm.mkdir()
    sysfs = build_tree(tmp_path / "sys", n_gpus=1)
    cfg = tmp_path / "config.json"
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"] += ["K3SAMD_VISIBLE_DEVICES=all",
                               "K3SAMD_INJECT_ROCM=1"]
    cfg.write_text(json.dumps(spec))
    subprocess.run(
        [str(RUNTIME), "--transform-only", str(cfg)],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs),
             "K3SAMD_DEV_ROOT": str(tmp_path / "nodev"),
             "K3SAMD_ROCM_ROOT": str(rocm)},
        check=True, capture_output=True, timeout=60)
    out = json.loads(cfg.read_text())
    rocm_mounts = [m for m in out["mounts"]
                   if m["destination"] == "/opt/rocm"]
    assert len(rocm_mounts) == 1
    assert rocm_mounts[0]["source"] == str(rocm)
    assert "ro" in rocm_mounts[0]["options"]


def test_rocm_default_with_env_optout(tmp_path):
    '''K3SAMD_INJECT_ROCM=0 overrides an inject-by-default runtime config.'''
    rocm = tmp_path / "opt-rocm"
    rocm.mkdir()
    sysfs = build_tree(tmp_path / "sys", n_gpus=1)
    cfg = tmp_path / "config.json"
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"] += ["K3SAMD_VISIBLE_DEVICES=all",
                               "K3SAMD_INJECT_ROCM=0"]
    cfg.write_text(json.dumps(spec))
    subprocess.run(
        [str(RUNTIME), "--transform-only", str(cfg)],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs),
             "K3SAMD_DEV_ROOT": str(tmp_path / "nodev"),
             "K3SAMD_ROCM_ROOT": str(rocm),
             "K3SAMD_INJECT_ROCM_DEFAULT": "1"},
        check=True, capture_output=True, timeout=60)
    out = json.loads(cfg.read_text())
    assert not any(m["destination"] == "/opt/rocm" for m in out["mounts"])
    # devices still injected
    assert "/dev/kfd" in device_paths(out)


def test_annotation_fallback_selection(tmp_path):
    '''With env scrubbed, the Allocate annotation still scopes injection.'''
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["annotations"] = {
        "k3samd.ai/allocated-gpus": "amdgpu-1a2b3c4d5e6f0001"}
    out, _ = transform(tmp_path, spec, n_gpus=4)
    assert device_paths(out) == ["/dev/kfd", "/dev/dri/renderD129",
                                 "/dev/dri/card1"]


def test_non_bmp_env_value_roundtrips_valid_utf8(tmp_path):
    """Emoji (non-BMP) in env values/annotations must come back as real
    UTF-8, whether escaped (\\ud83d\\ude00 surrogate pair -> one 4-byte
    code point) or raw — CESU-8 output would make runc reject the spec."""
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"].append("GREETING=hi \N{GRINNING FACE}")
    spec["annotations"] = {"note": "party \N{PARTY POPPER}"}
    sysfs = build_tree(tmp_path / "sys", n_gpus=1)
    cfg = tmp_path / "config.json"
    # write with escaped surrogates (ensure_ascii=True is json.dumps default)
    cfg.write_text(json.dumps(spec, ensure_ascii=True))
    subprocess.run(
        [str(RUNTIME), "--transform-only", str(cfg)],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs),
             "K3SAMD_DEV_ROOT": str(tmp_path / "nodev")},
        check=True, capture_output=True, timeout=60)
    raw = cfg.read_bytes()
    raw.decode("utf-8")  # must be valid UTF-8 (CESU-8 halves are not)
    out = json.loads(raw)
    assert "GREETING=hi \N{GRINNING FACE}" in out["process"]["env"]
    assert out["annotations"]["note"] == "party \N{PARTY POPPER}"


def test_cpu_only_node_injects_nothing(tmp_path):
    out, _ = transform(tmp_path, BASE_SPEC, n_gpus=0,
                       extra_env=["K3SAMD_VISIBLE_DEVICES=all"])
    assert "/dev/kfd" not in json.dumps(out)


def test_runc_passthrough_create(tmp_path):
    """Full runc-wrapper path: `create --bundle B id` transforms the
    bundle's config.json, then execs the (stubbed) real runc with
    identical arguments."""
    sysfs = build_tree(tmp_path / "sys", n_gpus=1)
    bundle = tmp_path / "bundle"
    bundle.mkdir()
    spec = json.loads(json.dumps(BASE_SPEC))
    spec["process"]["env"].append(
        "K3SAMD_VISIBLE_DEVICES=amdgpu-1a2b3c4d5e6f0000")
    (bundle / "config.json").write_text(json.dumps(spec))
    # stub runc that records its argv
    stub = tmp_path / "runc-stub"
    argv_log = tmp_path / "argv.txt"
    stub.write_text(f"#!/bin/sh\necho \"$@\" > {argv_log}\nexit 0\n")
    stub.chmod(0o755)
    proc = subprocess.run(
        [str(RUNTIME), "--root", "/run/x", "create",
         "--bundle", str(bundle), "ctr-1"],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs),
             "K3SAMD_DEV_ROOT": str(tmp_path / "nodev"),
             "K3SAMD_RUNC_PATH": str(stub)},
        capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr
    assert argv_log.read_text().split() == [
        "--root", "/run/x", "create", "--bundle", str(bundle), "ctr-1"]
    out = json.loads((bundle / "config.json").read_text())
    assert "/dev/kfd" in device_paths(out)


def test_runc_passthrough_noncreate_untouched(tmp_path):
    """`state`/`delete` etc. must pass through without touching configs."""
    stub = tmp_path / "runc-stub"
    stub.write_text("#!/bin/sh\nexit 0\n")
    stub.chmod(0o755)
    proc = subprocess.run(
        [str(RUNTIME), "state", "ctr-1"],
        env={"K3SAMD_RUNC_PATH": str(stub)},
        capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr


def test_malformed_config_fails(tmp_path):
    cfg = tmp_path / "config.json"
    cfg.write_text("{not json")
    proc = subprocess.run([str(RUNTIME), "--transform-only", str(cfg)],
                          capture_output=True, text=True, timeout=60)
    assert proc.returncode != 0

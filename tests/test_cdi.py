"""CDI path tests: spec generation + Allocate in --use-cdi mode."""

import json
import subprocess
from pathlib import Path

import grpc
import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree

REPO = Path(__file__).resolve().parent.parent
BIN = REPO / "native" / "bin"

IDENT = lambda b: b  # noqa: E731


def test_cdi_spec_generation(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=2)
    out = subprocess.run(
        [str(BIN / "k3samd-cdi-gen"), "--dev-root", str(tmp_path / "nodev")],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    spec = json.loads(out.stdout)
    assert spec["cdiVersion"] == "0.6.0"
    assert spec["kind"] == "amd.com/gpu"
    names = [d["name"] for d in spec["devices"]]
    # ordinal + stable-id per GPU, plus "all"
    assert "0" in names and "1" in names and "all" in names
    assert "amdgpu-1a2b3c4d5e6f0000" in names
    dev0 = next(d for d in spec["devices"] if d["name"] == "0")
    paths = [n["path"] for n in dev0["containerEdits"]["deviceNodes"]]
    assert paths == ["/dev/kfd", "/dev/dri/renderD128", "/dev/dri/card0"]
    all_dev = next(d for d in spec["devices"] if d["name"] == "all")
    assert len(all_dev["containerEdits"]["deviceNodes"]) == 1 + 2 * 2


def test_cdi_spec_file_output(tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=1)
    out_file = tmp_path / "amd.com-gpu.json"
    subprocess.run(
        [str(BIN / "k3samd-cdi-gen"), "--output", str(out_file)],
        env={"K3SAMD_SYSFS_ROOT": str(root)}, check=True, timeout=60)
    spec = json.loads(out_file.read_text())
    assert spec["kind"] == "amd.com/gpu"
    assert not (tmp_path / "amd.com-gpu.json.tmp").exists()


def test_allocate_cdi_mode(tmp_path, monkeypatch):
    # patch harness argv to include --use-cdi
    import test_deviceplugin as tdp
    orig_popen = tdp.subprocess.Popen

    def popen_with_cdi(argv, **kw):
        if str(tdp.PLUGIN) in argv[0]:
            argv = list(argv) + ["--use-cdi"]
        return orig_popen(argv, **kw)

    monkeypatch.setattr(tdp.subprocess, "Popen", popen_with_cdi)
    h = tdp.PluginHarness(tmp_path, n_gpus=2, replicas=1, register=False)
    try:
        devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
        resp = pb.decode_allocate_response(
            h.call("Allocate", pb.encode_allocate_request([[devs[0]["id"]]])))
        cr = resp[0]
        assert cr["devices"] == []  # CDI mode: no raw device specs
        raw = h.call("Allocate", pb.encode_allocate_request([[devs[0]["id"]]]))
        cdi_names = []
        for f, w, v in pb.fields(raw):
            if f == 1:
                for f2, _, v2 in pb.fields(v):
                    if f2 == 5:
                        for f3, _, v3 in pb.fields(v2):
                            if f3 == 1:
                                cdi_names.append(v3.decode())
        assert cdi_names == [f"amd.com/gpu={devs[0]['id']}"]
        assert cr["envs"]["K3SAMD_VISIBLE_DEVICES"] == devs[0]["id"]
    finally:
        h.close()

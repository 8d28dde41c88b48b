"""Compute-partition (CPX-style) topology: one physical GPU exposed as
multiple KFD nodes — same PCI BDF, distinct render nodes and unique_ids.
SURVEY.md §7 calls the node↔render↔BDF mapping under partitioning out as a
place where bugs live; pin the behavior: each partition is an independently
schedulable device whose Allocate injects its own render node (the shared
card node is resolved for both)."""

import json
import subprocess
from pathlib import Path

import pytest

import pb_v1beta1 as pb
from sysfs_builder import build_tree, MI355X_VRAM_BYTES
from test_deviceplugin import PluginHarness

REPO = Path(__file__).resolve().parent.parent
SMI = REPO / "native" / "bin" / "mi355x-smi"


def make_partitioned_tree(root):
    """2 partitions of one physical GPU: same BDF/card, render 128/129."""
    root = build_tree(root, n_gpus=1, n_cpu_nodes=1)
    import shutil
    nodes = root / "class/kfd/kfd/topology/nodes"
    # clone node 1 (the GPU) as node 2: second partition
    shutil.copytree(nodes / "1", nodes / "2")
    props = (nodes / "2" / "properties").read_text()
    props = props.replace("drm_render_minor 128", "drm_render_minor 129")
    props = props.replace(f"unique_id {0x1A2B3C4D5E6F0000}",
                          f"unique_id {0x1A2B3C4D5E6F0000 + 0x100}")
    (nodes / "2" / "properties").write_text(props)
    return root


def test_partitions_enumerate_independently(tmp_path):
    root = make_partitioned_tree(tmp_path / "sys")
    out = subprocess.run(
        [str(SMI), "--json"],
        env={"K3SAMD_SYSFS_ROOT": str(root), "PATH": "/usr/bin:/bin"},
        capture_output=True, text=True, timeout=60)
    j = json.loads(out.stdout)
    assert j["gpu_count"] == 2
    minors = sorted(g["render_minor"] for g in j["gpus"])
    assert minors == [128, 129]
    # both partitions share the physical card + BDF
    assert {g["pci_bdf"] for g in j["gpus"]} == {"0000:0c:00.0"}
    assert {g["card_index"] for g in j["gpus"]} == {0}
    assert len({g["id"] for g in j["gpus"]}) == 2  # ids stay distinct
    assert all(g["vram_bytes"] == MI355X_VRAM_BYTES for g in j["gpus"])


def test_partition_allocate_scopes_render_node(tmp_path):
    root = make_partitioned_tree(tmp_path / "sys")
    h = PluginHarness.__new__(PluginHarness)
    # build harness manually on the partitioned tree
    import time
    cfg = tmp_path / "config.yaml"
    cfg.write_text("version: v1\n")
    h.root = root
    h.plugin_sock = str(tmp_path / "amd.sock")
    h.kubelet = None
    h.proc = subprocess.Popen(
        [str(REPO / "native/bin/k3samd-device-plugin"), "--config", str(cfg),
         "--plugin-sock", h.plugin_sock, "--no-register"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    deadline = time.time() + 10
    while not Path(h.plugin_sock).exists():
        assert time.time() < deadline and h.proc.poll() is None
        time.sleep(0.05)
    import grpc
    h.channel = grpc.insecure_channel(f"unix:{h.plugin_sock}")
    try:
        devs = pb.decode_list_and_watch(next(h.stream("ListAndWatch")))
        assert len(devs) == 2
        # allocating ONE partition injects only its render node (plus the
        # shared kfd + card)
        resp = pb.decode_allocate_response(
            h.call("Allocate", pb.encode_allocate_request([[devs[1]["id"]]])))
        paths = sorted(d["host_path"] for d in resp[0]["devices"])
        assert paths == ["/dev/dri/card0", "/dev/dri/renderD129", "/dev/kfd"]
    finally:
        h.close()

"""Fixture-driven tests of the C++ KFD topology enumeration + mi355x-smi.

Mirrors the reference's smoke strategy (golden `nvidia-smi` output blocks,
/root/reference/README.md:71-93) as automated tests: the CPU-only node must
enumerate 0 GPUs (BASELINE.json config #1) and an 8xMI355X node must show
8 x 288 GiB gfx950 devices with 7 xGMI links each.
"""

import json
import subprocess

import pytest

from sysfs_builder import build_tree, MI355X_VRAM_BYTES


@pytest.fixture(scope="session")
def native_bin(tmp_path_factory):
    import pathlib
    root = pathlib.Path(__file__).resolve().parent.parent
    subprocess.run(["make", "-C", str(root / "native"), "-j8"], check=True,
                   capture_output=True)
    return root / "native" / "bin"


def run_smi(native_bin, sysfs_root, *args):
    proc = subprocess.run(
        [str(native_bin / "mi355x-smi"), *args],
        env={"K3SAMD_SYSFS_ROOT": str(sysfs_root), "PATH": "/usr/bin:/bin"},
        capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr
    return proc.stdout


def test_eight_gpu_node(native_bin, tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=8)
    out = json.loads(run_smi(native_bin, root, "--json"))
    assert out["gpu_count"] == 8
    assert out["driver_version"] == "6.12.12"
    minors = [g["render_minor"] for g in out["gpus"]]
    assert minors == list(range(128, 136))
    for i, g in enumerate(out["gpus"]):
        assert g["arch"] == "gfx950"
        assert g["name"] == "AMD Instinct MI355X"
        assert g["vram_bytes"] == MI355X_VRAM_BYTES
        assert g["compute_units"] == 256
        assert g["xgmi_links"] == 7
        assert g["card_index"] == i
        assert g["pci_bdf"] == f"0000:{0x0c + i:02x}:00.0"
        assert g["id"].startswith("amdgpu-1a2b3c4d5e6f")
        assert g["busy_percent"] == 0
        assert g["temp_milli_c"] == 53000
        assert g["power_uw"] == 135000000


def test_cpu_only_node(native_bin, tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=0)
    out = json.loads(run_smi(native_bin, root, "--json"))
    assert out["gpu_count"] == 0
    assert out["gpus"] == []


def test_no_kfd_at_all(native_bin, tmp_path):
    (tmp_path / "sys").mkdir()
    out = json.loads(run_smi(native_bin, tmp_path / "sys", "--json"))
    assert out["gpu_count"] == 0


def test_numa_mapping(native_bin, tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=4, n_cpu_nodes=2)
    out = json.loads(run_smi(native_bin, root, "--json"))
    # builder alternates CPU peers: gpu k -> cpu node (k % 2)
    assert [g["numa_node"] for g in out["gpus"]] == [0, 1, 0, 1]


def test_human_table(native_bin, tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=2)
    out = run_smi(native_bin, root)
    assert "AMD Instinct MI355X" in out
    assert "gfx950" in out
    assert "256" in out


def test_no_drm_tree(native_bin, tmp_path):
    '''KFD present but /sys/class/drm absent (minimal initramfs-style
    environments): enumeration still works, card unresolved, name falls
    back.'''
    root = build_tree(tmp_path / "sys", n_gpus=2, with_drm_cards=False)
    out = json.loads(run_smi(native_bin, root, "--json"))
    assert out["gpu_count"] == 2
    for g in out["gpus"]:
        assert g["card_index"] == -1
        assert g["render_minor"] >= 128
        assert g["name"]  # never empty


def test_unique_id_fallback_to_bdf(native_bin, tmp_path):
    root = build_tree(tmp_path / "sys", n_gpus=1)
    # remove unique_id from properties
    props = root / "class/kfd/kfd/topology/nodes/2/properties"
    lines = [l for l in props.read_text().splitlines()
             if not l.startswith("unique_id")]
    props.write_text("\n".join(lines) + "\n")
    out = json.loads(run_smi(native_bin, root, "--json"))
    assert out["gpus"][0]["id"] == "amdgpu-0000:0c:00.0"


def test_smi_health_columns(native_bin, tmp_path):
    """mi355x-smi surfaces the same RAS/error counters the device
    plugin's health model acts on (operator view == kubelet view)."""
    import json
    import subprocess
    from sysfs_builder import make_gpu_sick
    root = build_tree(tmp_path / "sys", n_gpus=2)
    make_gpu_sick(root, 1, umc_ue=3, umc_ce=17, resets=1)
    env = {"K3SAMD_SYSFS_ROOT": str(root)}
    out = subprocess.run([str(native_bin / "mi355x-smi")], env=env,
                         capture_output=True, text=True, timeout=60)
    assert "ECC ue/ce" in out.stdout
    assert "3/17" in out.stdout       # the sick GPU's counters
    assert "n/a" in out.stdout        # GPU 0 has no RAS sysfs -> n/a
    j = json.loads(subprocess.run(
        [str(native_bin / "mi355x-smi"), "--json"], env=env,
        capture_output=True, text=True, timeout=60).stdout)
    sick = j["gpus"][1]
    assert sick["ras_supported"] == "true"
    assert sick["ras_uncorrectable"] == 3
    assert sick["ras_correctable"] == 17
    assert sick["reset_count"] == 1
    healthy = j["gpus"][0]
    assert healthy["ras_supported"] == "false"
    assert healthy["ras_uncorrectable"] == -1


def test_smi_topo_matrix(native_bin, tmp_path):
    """`mi355x-smi --topo` — the nvidia-smi `topo -m` analog: pairwise
    xGMI connectivity from KFD io_links. The 4-GPU fixture is fully
    connected; a 1-GPU tree has no peers."""
    root = build_tree(tmp_path / "sys", n_gpus=4)
    out = run_smi(native_bin, root, "--topo")
    lines = [ln for ln in out.splitlines() if ln.strip().startswith("GPU")]
    assert len(lines) == 4 + 0 or len(lines) >= 4
    rows = [ln.split() for ln in out.splitlines()
            if ln.strip().startswith("GPU") and "NUMA" not in ln]
    assert len(rows) == 4
    for i, row in enumerate(rows):
        cells = row[1:5]
        assert cells[i] == "X"
        for j, c in enumerate(cells):
            if j != i:
                assert c == "XGMI", (i, j, row)
    assert "0000:0c:00.0" in out  # BDF column
    # single-GPU tree: no xGMI peers -> matrix is just the self cell
    root1 = build_tree(tmp_path / "sys1", n_gpus=1)
    out1 = run_smi(native_bin, root1, "--topo")
    row = [ln for ln in out1.splitlines() if ln.strip().startswith("GPU0")][0]
    assert "XGMI" not in row

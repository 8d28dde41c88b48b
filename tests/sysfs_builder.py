"""Synthesize KFD sysfs topology fixture trees for CPU-only testing.

Models the layout of /sys on an MI355X node (and a CPU-only node) closely
enough for native/topology/kfd_topology.cc: nodes with `properties`,
`mem_banks`, `io_links`, plus /sys/class/drm card devices for runtime stats.
"""

from __future__ import annotations

from pathlib import Path

MI355X_VRAM_BYTES = 288 * (1 << 30)  # 288 GiB HBM3E
MI355X_DEVICE_ID = 0x75A0
AMD_VENDOR_ID = 0x1002
GFX950_TARGET_VERSION = 90500
IOLINK_PCIE = 2
IOLINK_XGMI = 11


def _write_props(path: Path, props: dict) -> None:
    path.parent.mkdir(parents=True, exist_ok=True)
    path.write_text("".join(f"{k} {v}\n" for k, v in props.items()))


def build_tree(
    root: Path,
    n_gpus: int = 8,
    n_cpu_nodes: int = 2,
    vram_bytes: int = MI355X_VRAM_BYTES,
    with_drm_cards: bool = True,
    driver_version: str = "6.12.12",
    busy_percent: int = 0,
    vram_used: int = 2 * (1 << 20),
) -> Path:
    """Create a sysfs-like tree under `root`; returns `root`."""
    root = Path(root)
    (root / "module/amdgpu").mkdir(parents=True, exist_ok=True)
    if driver_version:
        (root / "module/amdgpu/version").write_text(driver_version + "\n")

    nodes = root / "class/kfd/kfd/topology/nodes"

    # CPU nodes first (KFD enumerates CPUs and GPUs in one namespace)
    for i in range(n_cpu_nodes):
        _write_props(nodes / str(i) / "properties", {
            "cpu_cores_count": 96,
            "simd_count": 0,
            "mem_banks_count": 1,
            "io_links_count": 0,
            "vendor_id": 0,
            "device_id": 0,
        })
        (nodes / str(i) / "name").write_text("\n")

    for k in range(n_gpus):
        n = n_cpu_nodes + k
        bus = 0x0C + k
        location_id = bus << 8
        ndir = nodes / str(n)
        _write_props(ndir / "properties", {
            "cpu_cores_count": 0,
            "simd_count": 1024,          # 256 CUs x 4 SIMDs
            "simd_per_cu": 4,
            "max_waves_per_simd": 8,
            "lds_size_in_kb": 160,
            "wave_front_size": 64,
            "gfx_target_version": GFX950_TARGET_VERSION,
            "vendor_id": AMD_VENDOR_ID,
            "device_id": MI355X_DEVICE_ID,
            "unique_id": 0x1A2B3C4D5E6F0000 + k,
            "domain": 0,
            "location_id": location_id,
            "drm_render_minor": 128 + k,
            "num_xcc": 8,
        })
        (ndir / "name").write_text("AMD Instinct MI355X\n")

        _write_props(ndir / "mem_banks/0/properties", {
            "heap_type": 1,  # FB public
            "size_in_bytes": vram_bytes,
            "flags": 0,
        })

        # io_links: one PCIe link to the nearest CPU node + 7 xGMI peers
        link = 0
        cpu_peer = k % max(n_cpu_nodes, 1) if n_cpu_nodes else 0
        _write_props(ndir / f"io_links/{link}/properties", {
            "type": IOLINK_PCIE,
            "node_from": n,
            "node_to": cpu_peer,
            "weight": 20,
        })
        link += 1
        for peer in range(n_gpus):
            if peer == k:
                continue
            _write_props(ndir / f"io_links/{link}/properties", {
                "type": IOLINK_XGMI,
                "node_from": n,
                "node_to": n_cpu_nodes + peer,
                "weight": 15,
                "min_bandwidth": 153000,
                "max_bandwidth": 153000,
            })
            link += 1

        if with_drm_cards:
            card = root / f"class/drm/card{k}/device"
            card.mkdir(parents=True, exist_ok=True)
            bdf = f"0000:{bus:02x}:00.0"
            (card / "uevent").write_text(
                f"DRIVER=amdgpu\nPCI_SLOT_NAME={bdf}\n")
            (card / "mem_info_vram_total").write_text(f"{vram_bytes}\n")
            (card / "mem_info_vram_used").write_text(f"{vram_used}\n")
            (card / "gpu_busy_percent").write_text(f"{busy_percent}\n")
            hw = card / "hwmon/hwmon0"
            hw.mkdir(parents=True, exist_ok=True)
            (hw / "temp1_input").write_text("53000\n")
            (hw / "power1_average").write_text("135000000\n")  # 135 W

    return root


def make_gpu_sick(root, card_index, *, umc_ue=0, umc_ce=0, gfx_ue=0,
                  gfx_ce=0, pcie_replays=None, resets=None, aca=False,
                  umc_de=0, fatal_events=None, bad_pages=None):
    """Write amdgpu RAS/error-state files for a present-but-sick GPU
    (native/topology/gpu_health.cc reads these). Two generations, both
    verified against real sysfs:
      legacy (aca=False): `ras/<block>_err_count` with "ue: N\\nce: N"
      ACA    (aca=True):  `ras/aca_<block>` with "ue: N\\nce: N\\nde: N"
                          (the MI355X format, gpurun_out/r2_aca_probe.txt)
    plus `ras/event_state`, `ras/gpu_vram_bad_pages`,
    `pcie_replay_count`, `reset_count`. Re-call with zeros to 'heal'."""
    dev = Path(root) / f"class/drm/card{card_index}/device"
    ras = dev / "ras"
    ras.mkdir(parents=True, exist_ok=True)
    if aca:
        (ras / "aca_umc").write_text(
            f"ue: {umc_ue}\nce: {umc_ce}\nde: {umc_de}\n")
        (ras / "aca_gfx").write_text(
            f"ue: {gfx_ue}\nce: {gfx_ce}\nde: 0\n")
    else:
        (ras / "umc_err_count").write_text(f"ue: {umc_ue}\nce: {umc_ce}\n")
        (ras / "gfx_err_count").write_text(f"ue: {gfx_ue}\nce: {gfx_ce}\n")
    if fatal_events is not None:
        (ras / "event_state").write_text(
            "current seqno: 7\n"
            f"Fatal Error: count:{fatal_events}, last_seqno:0\n"
            "Poison Creation: count:0, last_seqno:0\n"
            "Poison Consumption: count:0, last_seqno:0\n")
    if bad_pages is not None:
        rows = "".join(f"0x{0x1000 * (i + 1):012x} : 0x1000 : R\n"
                       for i in range(bad_pages))
        (ras / "gpu_vram_bad_pages").write_text(rows)
    if pcie_replays is not None:
        (dev / "pcie_replay_count").write_text(f"{pcie_replays}\n")
    if resets is not None:
        (dev / "reset_count").write_text(f"{resets}\n")

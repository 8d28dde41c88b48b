"""GPU tests for the standalone in-pod payload binaries (mi-stream,
mi-allreduce) — the executables the smoke manifests run where the
reference ran `nvidia-smi` (/root/reference/nvidia-smi.yaml:13)."""

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
def test_mi_stream_runs():
    proc = subprocess.run([str(BIN / "mi-stream"), "--mib", "512",
                           "--iters", "5"],
                          capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr
    # built-in numerics oracle must have run and passed (exit 3 otherwise)
    assert "numerics: triad exact" in proc.stdout
    assert "MX-fp8 identity exact" in proc.stdout
    last = proc.stdout.strip().splitlines()[-1]
    j = json.loads(last)
    assert j["payload"] == "mi-stream"
    assert j["arch"].startswith("gfx950")
    # >L3 buffers on MI355X must stream well past 4 TB/s
    assert j["triad_gbps"] > 4000, j
    assert j["mfma_bf16_tflops"] > 500, j


@requires_gpu
def test_mi_stream_all_gpus():
    proc = subprocess.run([str(BIN / "mi-stream"), "--all-gpus",
                           "--mib", "256", "--iters", "5"],
                          capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr
    last = proc.stdout.strip().splitlines()[-1]
    j = json.loads(last)
    assert j["mode"] == "all-gpus"
    assert j["n_gpus"] >= 1
    assert j["aggregate_triad_gbps"] > 4000 * j["n_gpus"]


@requires_gpu
def test_mi_stream_burn_short():
    proc = subprocess.run([str(BIN / "mi-stream"), "--burn", "4",
                           "--mib", "256"],
                          capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr
    j = json.loads(proc.stdout.strip().splitlines()[-1])
    assert j["payload"] == "mi-burn"
    assert j["avg_triad_gbps"] > 2000  # concurrent with MFMA load
    assert j["avg_mfma_tflops"] > 50


@requires_gpu
def test_mi_allreduce_single_gpu():
    proc = subprocess.run([str(BIN / "mi-allreduce"), "--ngpus", "1",
                           "--min-mib", "4", "--max-mib", "16",
                           "--iters", "5"],
                          capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr
    last = proc.stdout.strip().splitlines()[-1]
    j = json.loads(last)
    assert j["payload"] == "mi-allreduce"
    assert j["n_gpus"] == 1


@requires_gpu
def test_mi355x_smi_real_sysfs():
    """mi355x-smi against the real /sys of the GPU box."""
    proc = subprocess.run([str(BIN / "mi355x-smi"), "--json"],
                          capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr
    j = json.loads(proc.stdout)
    assert j["gpu_count"] >= 1
    g = j["gpus"][0]
    assert g["arch"] == "gfx950"
    assert g["vram_bytes"] > 200 * (1 << 30)
    assert g["render_minor"] >= 128
    # health/RAS fields present (values depend on the box's driver build;
    # -1 = not exposed is acceptable, absence is not)
    for key in ("ras_supported", "ras_uncorrectable", "ras_correctable",
                "ras_deferred", "fatal_ras_events", "retired_vram_pages",
                "pcie_replay_count", "reset_count"):
        assert key in g, key
    if g["ras_supported"] == "true":
        # the MI355X pool exposes ACA RAS banks: counters must be real
        # numbers (not the -1 'absent' sentinel) once RAS is present
        assert g["ras_uncorrectable"] >= 0
        assert g["ras_correctable"] >= 0
        assert g["ras_deferred"] >= 0


@pytest.mark.gpu
def test_mi355x_smi_topo_real():
    """--topo connectivity matrix renders on the real box (single-GPU
    lease: one row, no peer columns marked XGMI unless peers are
    visible KFD nodes)."""
    proc = subprocess.run([str(BIN / "mi355x-smi"), "--topo"],
                          capture_output=True, text=True, timeout=60)
    assert proc.returncode == 0, proc.stderr
    assert "GPU0" in proc.stdout
    assert "Legend" in proc.stdout

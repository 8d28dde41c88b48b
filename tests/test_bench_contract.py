"""CPU-side checks of the bench.py driver contract."""

import json
import subprocess
import sys

import bench


def test_build_result_schema():
    res = bench.build_result(value=12345.678, n_gpus=8, steps=200, warmup=50,
                             ms_per_step=0.51, buffer_mib=1024,
                             variant="plain")
    # required contract keys
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in res
    assert res["metric"] == "hip_stream_triad_gbps"
    assert res["higher_is_better"] is True
    assert res["scaling"] == "weak"
    assert res["data"] == "synthetic"
    assert res["config"]["parallelism"] == "dp8"
    json.dumps(res)  # serializable


def test_default_args():
    args = bench.parse_args([])
    assert args.steps == 200 and args.warmup == 50
    assert args.buffer_mib == 1024


def test_bench_fails_cleanly_without_gpu(repo_root):
    """On a CPU-only box bench.py must exit non-zero with a clear message."""
    import torch
    if torch.cuda.is_available():
        return  # covered by the gpu bench run instead
    proc = subprocess.run(
        [sys.executable, str(repo_root / "bench.py"), "--steps", "1",
         "--warmup", "0"],
        capture_output=True, text=True, timeout=300)
    assert proc.returncode != 0
    assert "GPU" in proc.stderr

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


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _torchrun_rehearsal(repo_root, nproc, timeout=600):
    """Launch bench.py exactly the way the driver does
    (torch.distributed.run, one rank per 'GPU'), in CPU-rehearsal mode
    (gloo). This rehearses the rendezvous, barriers, max-over-ranks
    reduction and the JSON contract at the real world width."""
    import os
    env = dict(os.environ)
    env["K3SAMD_BENCH_REHEARSAL"] = "1"
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", f"--nproc-per-node={nproc}",
         "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
         str(repo_root / "bench.py"),
         "--gpus", str(nproc), "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=timeout, env=env,
        cwd=str(repo_root))
    assert proc.returncode == 0, proc.stderr[-3000:]
    lines = [ln for ln in proc.stdout.splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, proc.stdout  # exactly ONE JSON line (rank 0)
    return json.loads(lines[0])


def test_torchrun_rehearsal_world4(repo_root):
    res = _torchrun_rehearsal(repo_root, 4)
    assert res["n_gpus"] == 4
    assert res["rehearsal"] is True  # can never pass as a measurement
    assert res["config"]["parallelism"] == "dp4"
    assert res["scaling"] == "weak"


def test_torchrun_rehearsal_world8(repo_root):
    """The exact width the driver's 8-GPU scale run will use."""
    res = _torchrun_rehearsal(repo_root, 8)
    assert res["n_gpus"] == 8
    assert res["rehearsal"] is True
    assert res["config"]["parallelism"] == "dp8"


def test_duplicate_device_binding_detected():
    """Unit test of the physical-GPU duplicate detector bench.py runs
    before every multi-GPU measurement."""
    ok = [("host", 0, 0xc1, 0), ("host", 0, 0xc5, 0), ("host", 0, 0xc9, 0)]
    assert bench.canonical_device_ids(ok) == []
    dup = [("host", 0, 0xc1, 0), ("host", 0, 0xc1, 0), ("host", 0, 0xc9, 0)]
    found = bench.canonical_device_ids(dup)
    assert len(found) == 1 and found[0][:2] == (0, 1)
    # same BDF on different hosts is NOT a duplicate
    multi = [("hostA", 0, 0xc1, 0), ("hostB", 0, 0xc1, 0)]
    assert bench.canonical_device_ids(multi) == []


def test_rccl_transport_log_parsing(tmp_path):
    log = tmp_path / "rccl.log"
    log.write_text(
        "node:123:145 [0] NCCL INFO Channel 00/0 : 0[0] -> 1[1] via P2P/IPC\n"
        "node:123:145 [0] NCCL INFO Channel 01/0 : 0[0] -> 1[1] via P2P/IPC/read\n"
        "node:123:146 [1] NCCL INFO Channel 00/0 : 1[1] -> 0[0] via SHM/direct/direct\n"
        "node:123:146 [1] NCCL INFO Comm config Blocking set to 1\n")
    counts = bench.parse_rccl_transports(str(log))
    assert counts == {"P2P": 2, "SHM": 1}
    assert bench.parse_rccl_transports(str(tmp_path / "missing.log")) == {}

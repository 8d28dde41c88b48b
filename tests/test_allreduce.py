"""Distributed-path tests: gloo backend, world_size=2, on CPU.

Validates the same code path the RCCL (nccl-backend) GPU sweep uses —
correctness of the collective plus the busbw bookkeeping — so the 8-GPU
run at round end is correct by construction.
"""

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def free_port() -> str:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return str(port)

WORKER = r"""
import json, os, sys
import torch, torch.distributed as dist
sys.path.insert(0, os.environ["K3_REPO"])
from k3samd.parallel.allreduce import busbw_sweep

dist.init_process_group(backend="gloo",
                        rank=int(os.environ["RANK"]),
                        world_size=int(os.environ["WORLD_SIZE"]))
rank = dist.get_rank()

# correctness: sum of per-rank constants
t = torch.full((1024,), float(rank + 1))
dist.all_reduce(t)
assert torch.equal(t, torch.full((1024,), 3.0)), t[0]

rows = busbw_sweep([1 << 16, 1 << 18], iters=4)
assert len(rows) == 2
assert all(r["busbw_gbps"] > 0 for r in rows)
# world=2: busbw = algbw * 2*(2-1)/2 = algbw
for r in rows:
    assert abs(r["busbw_gbps"] - r["algbw_gbps"]) < 1e-6 + 0.02 * r["algbw_gbps"]

# other collectives run and produce sane numbers on gloo too
for op in ("all_gather", "broadcast"):
    rows = busbw_sweep([1 << 18], iters=2, op=op)
    assert rows[0]["bytes"] == 1 << 18
    assert rows[0]["busbw_gbps"] > 0, (op, rows)
if rank == 0:
    print("WORKER_OK", json.dumps(rows))
dist.destroy_process_group()
"""


def test_gloo_world2(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    procs = []
    port = free_port()
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port,
            "K3_REPO": str(REPO),
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=300) for p in procs]
    for p, (out, err) in zip(procs, outs):
        assert p.returncode == 0, err
    assert "WORKER_OK" in outs[0][0]


def test_cli_single_rank():
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": free_port()})
    proc = subprocess.run(
        [sys.executable, "-m", "k3samd.parallel.allreduce",
         "--backend", "gloo", "--min-mib", "1", "--max-mib", "1",
         "--iters", "2"],
        cwd=str(REPO), env=env, capture_output=True, text=True, timeout=300)
    assert proc.returncode == 0, proc.stderr
    last = proc.stdout.strip().splitlines()[-1]
    j = json.loads(last)
    assert j["backend"] == "gloo" and j["world_size"] == 1


def test_torchrun_world4_cli(tmp_path):
    """The exact in-pod launch form (torch.distributed.run, the module
    CLI) rehearsed at width 4 on gloo — the same code path the 8-GPU
    RCCL pod runs, minus the backend."""
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node=4",
         "--master-addr", "127.0.0.1", "--master-port", free_port(),
         "-m", "k3samd.parallel.allreduce",
         "--backend", "gloo", "--min-mib", "1", "--max-mib", "1",
         "--iters", "2"],
        cwd=str(REPO), env=dict(os.environ), capture_output=True, text=True,
        timeout=600)
    assert proc.returncode == 0, proc.stderr[-3000:]
    last = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(last) == 1
    j = json.loads(last[0])
    assert j["world_size"] == 4
    assert j["rows"][0]["busbw_gbps"] > 0


def test_verify_flag_in_output(tmp_path):
    """The payload's JSON carries the self-verification verdict (rank r
    contributes r+1; every element must equal world*(world+1)/2)."""
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node=2",
         "--master-addr", "127.0.0.1", "--master-port", free_port(),
         "-m", "k3samd.parallel.allreduce",
         "--backend", "gloo", "--min-mib", "1", "--max-mib", "1",
         "--iters", "2"],
        cwd=str(REPO), env=dict(os.environ), capture_output=True, text=True,
        timeout=600)
    assert proc.returncode == 0, proc.stderr[-2000:]
    j = json.loads([ln for ln in proc.stdout.splitlines()
                    if ln.startswith("{")][0])
    assert j["verify_ok"] is True

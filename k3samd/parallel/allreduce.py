"""RCCL-over-xGMI collective smoke path (torch.distributed).

The multi-process counterpart of native/hipsmoke/mi_allreduce.hip: one rank
per GPU over the "nccl" backend (RCCL on ROCm), or "gloo" on CPU where the
same code doubles as the distributed-path unit test.

Run inside a pod requesting `amd.com/gpu: N`:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 -m k3samd.parallel.allreduce
"""

from __future__ import annotations

import argparse
import json
import os
import time


# busbw normalization factors (bytes actually crossing links per byte of
# payload, ring-equivalent — the standard nccl-tests convention)
_BUSBW_FACTOR = {
    "all_reduce": lambda n: 2 * (n - 1) / n,
    "all_gather": lambda n: (n - 1) / n,
    "reduce_scatter": lambda n: (n - 1) / n,
    "broadcast": lambda n: 1.0,
}


def _run_collective(op, t, world, device, group):
    import torch
    import torch.distributed as dist

    if op == "all_reduce":
        dist.all_reduce(t, group=group)
    elif op == "broadcast":
        dist.broadcast(t, src=0, group=group)
    elif op == "all_gather":
        out = torch.empty(t.numel() * world, dtype=t.dtype, device=t.device)
        dist.all_gather_into_tensor(out, t, group=group)
    elif op == "reduce_scatter":
        inp = torch.empty(t.numel() * world, dtype=t.dtype, device=t.device)
        dist.reduce_scatter_tensor(t, inp, group=group)
    else:
        raise ValueError(f"unknown op {op}")


def busbw_sweep(sizes_bytes, iters=20, device=None, group=None,
                op="all_reduce"):
    """Run `op` over each size `iters` times; returns
    [{bytes, algbw, busbw}]. busbw uses the standard ring-equivalent
    normalization so values are comparable across world sizes and ops."""
    import torch
    import torch.distributed as dist

    world = dist.get_world_size(group)
    factor = _BUSBW_FACTOR[op](world)
    rows = []
    for bytes_ in sizes_bytes:
        n = max(int(bytes_) // 4, 1)
        t = torch.rand(n, dtype=torch.float32, device=device)
        for _ in range(3):
            _run_collective(op, t, world, device, group)
        if device is not None and str(device).startswith("cuda"):
            torch.cuda.synchronize(device)
        dist.barrier(group)
        t0 = time.perf_counter()
        for _ in range(iters):
            _run_collective(op, t, world, device, group)
        if device is not None and str(device).startswith("cuda"):
            torch.cuda.synchronize(device)
        elapsed = time.perf_counter() - t0
        # max over ranks so the report reflects the slowest participant
        e = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX, group=group)
        per_op = float(e[0]) / iters
        algbw = n * 4 / per_op / 1e9
        busbw = algbw * factor
        rows.append({"bytes": n * 4, "algbw_gbps": round(algbw, 4),
                     "busbw_gbps": round(busbw, 4)})
    return rows


def main(argv=None) -> int:
    import torch
    import torch.distributed as dist

    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--min-mib", type=int, default=1)
    p.add_argument("--max-mib", type=int, default=1024)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--backend", default="auto")
    p.add_argument("--op", default="all_reduce",
                   choices=sorted(_BUSBW_FACTOR))
    args = p.parse_args(argv)

    backend = args.backend
    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    rccl_log = None
    if backend == "nccl" and world > 1:
        # capture RCCL's channel-setup log so the run can prove (or
        # refute) xGMI p2p transport selection on its own
        from k3samd.utils.rccl_diag import enable_rccl_debug_capture
        rccl_log = enable_rccl_debug_capture()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29572")
    dist.init_process_group(backend=backend, rank=rank, world_size=world)

    device = None
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
        if world > 1:
            # two ranks on one physical GPU would corrupt the busbw claim
            from k3samd.utils.rccl_diag import assert_unique_device_binding
            assert_unique_device_binding(dist, device, rank)

    sizes = []
    b = args.min_mib << 20
    while b <= args.max_mib << 20:
        sizes.append(b)
        b *= 4
    rows = busbw_sweep(sizes, iters=args.iters, device=device, op=args.op)

    # correctness verdict: rank r contributes (r+1); every element must be
    # world*(world+1)/2 on every rank
    vt = torch.full((1 << 12,), float(rank + 1),
                    device=device if device is not None else "cpu")
    dist.all_reduce(vt)
    expect = world * (world + 1) / 2.0
    verified = bool((vt == expect).all().item())
    vflag = torch.tensor([1 if verified else 0],
                         device=device if device is not None else "cpu")
    dist.all_reduce(vflag, op=dist.ReduceOp.MIN)
    verified = bool(int(vflag[0]) == 1)

    transports = None
    if rccl_log is not None:
        from k3samd.utils.rccl_diag import gather_transport_counts
        transports = gather_transport_counts(dist, rccl_log, rank, world)

    if rank == 0:
        for r in rows:
            print(f"{r['bytes']:>12d} B  algbw {r['algbw_gbps']:8.2f} GB/s"
                  f"  busbw {r['busbw_gbps']:8.2f} GB/s")
        result = {"payload": "allreduce", "backend": backend,
                  "op": args.op, "world_size": world, "rows": rows,
                  "verify_ok": verified}
        if backend == "nccl" and world > 1:
            # xGMI sanity verdict: the fully-connected
            # MI355X node must beat one xGMI link's ~153 GB/s at large
            # sizes, and every RCCL channel should be P2P (SHM = host
            # bounce = the injected /dev/dri set is broken)
            best = max(r["busbw_gbps"] for r in rows)
            result["max_busbw_gbps"] = best
            result["xgmi_p2p_ok"] = best > 153.0
            if transports is not None:
                result["rccl_transports"] = transports
                non_p2p = sum(v for k, v in transports.items() if k != "P2P")
                result["rccl_all_p2p"] = bool(transports.get("P2P")) \
                    and non_p2p == 0
        print(json.dumps(result))
    dist.destroy_process_group()
    return 0 if verified else 4


if __name__ == "__main__":
    raise SystemExit(main())

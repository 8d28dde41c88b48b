"""Multi-GPU run self-diagnosis helpers.

Used by bench.py and k3samd.parallel.allreduce so the FIRST real 8-GPU
contact is a measurement, not a debug session:

  * enable_rccl_debug_capture(): route RCCL's INFO log to a per-process
    file before communicator creation, so the run can prove which
    transport (xGMI P2P vs SHM host-bounce vs NET) each channel used.
  * parse_rccl_transports(): count channel-setup lines by transport.
  * assert_unique_device_binding(): fail loudly if two ranks are bound
    to the same physical GPU (host + PCI BDF identity) — the silent
    foot-gun that halves an aggregate-bandwidth claim.
"""

from __future__ import annotations

import os


def canonical_device_ids(entries):
    """entries: per-rank (host, pci_domain, pci_bus, pci_device) tuples.
    Returns [(rank_a, rank_b, ident)] duplicates (empty = binding OK)."""
    seen = {}
    dups = []
    for rank, ident in enumerate(entries):
        ident = tuple(ident)
        if ident in seen:
            dups.append((seen[ident], rank, ident))
        else:
            seen[ident] = rank
    return dups


def assert_unique_device_binding(dist, device, rank):
    """Gather every rank's bound-GPU identity and raise if any physical
    GPU appears twice. Returns the gathered identity list."""
    import socket as _socket
    import torch
    props = torch.cuda.get_device_properties(device)
    ident = (_socket.gethostname(), props.pci_domain_id, props.pci_bus_id,
             props.pci_device_id)
    gathered = [None] * dist.get_world_size()
    dist.all_gather_object(gathered, ident)
    dups = canonical_device_ids(gathered)
    if dups:
        msg = "; ".join(f"ranks {a} and {b} both bound to PCI "
                        f"{i[1]:04x}:{i[2]:02x}:{i[3]:02x} on {i[0]}"
                        for a, b, i in dups)
        raise RuntimeError(f"device binding violation: {msg}")
    return gathered


def enable_rccl_debug_capture():
    """Must run BEFORE the first collective creates the communicator.
    Honors pre-set NCCL_DEBUG/NCCL_DEBUG_FILE; returns the log path."""
    import tempfile
    log_dir = os.environ.get("K3SAMD_RCCL_LOG_DIR", tempfile.gettempdir())
    path = os.path.join(log_dir, f"rccl_debug_{os.getpid()}.log")
    os.environ.setdefault("NCCL_DEBUG", "INFO")
    os.environ.setdefault("NCCL_DEBUG_FILE", path)
    return os.environ["NCCL_DEBUG_FILE"]


def parse_rccl_transports(log_path):
    """Count channel-setup lines by transport from an RCCL debug log:
    'via P2P' (xGMI direct), 'via SHM' (host bounce — wrong on a single
    node), 'via NET'. Returns {} when the log is missing/empty."""
    counts = {}
    try:
        with open(log_path, errors="replace") as f:
            for line in f:
                if "->" not in line or " via " not in line:
                    continue
                transport = line.split(" via ", 1)[1].split()[0].split("/")[0]
                counts[transport] = counts.get(transport, 0) + 1
    except OSError:
        return {}
    return counts


def gather_transport_counts(dist, rccl_log, rank, world):
    """All-gather per-rank transport counts; returns the summed dict on
    rank 0, None elsewhere (or None when no log was captured)."""
    mine = parse_rccl_transports(rccl_log) if rccl_log else {}
    gathered = [None] * world
    dist.all_gather_object(gathered, mine)
    if rank != 0:
        return None
    total = {}
    for g in gathered:
        for k, v in (g or {}).items():
            total[k] = total.get(k, 0) + v
    return total

"""Multi-node health probes for the torchrun/LWS deployment contract.

Reference parity: python/huggingfaceserver multinode health_check.py:25-194
— Ray-cluster startup/readiness/liveness probes wired into the
kserve-huggingfaceserver-multinode runtime. Ours replaces Ray with the
torchrun rendezvous (RANK/WORLD_SIZE/MASTER_ADDR env, the same contract
the llmisvc LWS workload renders), so the probes check:

- ``probe_master``: the rendezvous TCP endpoint accepts connections
  (worker startup probe — don't start ranks before the leader listens)
- ``probe_store``: a TCPStore round-trip through the master (readiness —
  the leader's store answers; workers count themselves in)
- ``probe_gpu``: the local GPU answers a trivial kernel (liveness)

CLI: ``python -m kserve_amd.parallel.health <probe> [--timeout S]`` exits
0/1 for exec-probe use in pod specs.
"""

from __future__ import annotations

import os
import socket
import sys
import time


def probe_master(host: str, port: int, timeout_s: float = 2.0) -> bool:
    """TCP connect to the rendezvous endpoint."""
    try:
        with socket.create_connection((host, port), timeout=timeout_s):
            return True
    except OSError:
        return False


def probe_store(
    host: str,
    port: int,
    rank: int,
    world_size: int,
    timeout_s: float = 5.0,
) -> bool:
    """TCPStore round-trip: the leader hosts the store; any rank can set
    and read back its own health key."""
    import datetime

    import torch.distributed as dist

    try:
        store = dist.TCPStore(
            host,
            port,
            world_size,
            is_master=False,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
        key = f"health-{rank}-{int(time.time())}"
        store.set(key, b"ok")
        return store.get(key) == b"ok"
    except Exception:
        return False


def probe_gpu(timeout_s: float = 10.0) -> bool:
    """A trivial kernel answers on the local device (liveness)."""
    try:
        import torch

        if not torch.cuda.is_available():
            return False
        x = torch.ones(8, device="cuda")
        y = (x + x).sum()
        torch.cuda.synchronize()
        return float(y) == 16.0
    except Exception:
        return False


def main(argv=None) -> int:
    import argparse

    ap = argparse.ArgumentParser(description="multinode health probes")
    ap.add_argument("probe", choices=["master", "store", "gpu"])
    ap.add_argument("--timeout", type=float, default=5.0)
    ap.add_argument("--master-addr",
                    default=os.environ.get("MASTER_ADDR", "127.0.0.1"))
    # the store probe uses the rendezvous port + 1 by convention so it
    # never interferes with an in-progress process-group init
    ap.add_argument("--master-port", type=int,
                    default=int(os.environ.get("MASTER_PORT", "29500")))
    ap.add_argument("--rank", type=int,
                    default=int(os.environ.get("RANK", "0")))
    ap.add_argument("--world-size", type=int,
                    default=int(os.environ.get("WORLD_SIZE", "1")))
    args = ap.parse_args(argv)
    if args.probe == "master":
        ok = probe_master(args.master_addr, args.master_port, args.timeout)
    elif args.probe == "store":
        ok = probe_store(
            args.master_addr, args.master_port, args.rank, args.world_size,
            args.timeout,
        )
    else:
        ok = probe_gpu(args.timeout)
    print("ok" if ok else "unhealthy")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())

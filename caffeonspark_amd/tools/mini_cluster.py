"""Standalone no-Spark cluster launcher (reference: caffe_mini_cluster,
SURVEY.md §3.4): spawns one training process per GPU on this node,
rendezvous over 127.0.0.1, each rank running the CaffeOnSpark train path
with RCCL DistributedSync.

  python -m caffeonspark_amd.tools.mini_cluster -cluster 8 \
      -conf solver.prototxt -train

SIGINT/SIGTERM snapshot-and-stop (reference signal handler,
caffe_mini_cluster.cpp:233-240).
"""

from __future__ import annotations

import os
import signal
import sys
from typing import List, Optional

import torch
import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int, argv: List[str]):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from ..api import main as cos_main

    def handler(signum, frame):
        global _proc
        import caffeonspark_amd.data.processor as pmod
        inst = pmod._instance
        if inst is not None:
            if rank == 0:
                try:
                    inst.snapshot()
                except Exception:
                    pass
            inst.stop_flag.set()
        sys.exit(0)

    signal.signal(signal.SIGINT, handler)
    signal.signal(signal.SIGTERM, handler)
    cos_main(argv)


def main(argv: Optional[List[str]] = None) -> None:
    argv = list(argv if argv is not None else sys.argv[1:])
    cluster = 1
    if "-cluster" in argv:
        i = argv.index("-cluster")
        cluster = int(argv[i + 1])
        del argv[i:i + 2]
    elif torch.cuda.is_available():
        cluster = torch.cuda.device_count()
    port = int(os.environ.get("MASTER_PORT", "29603"))
    if cluster <= 1:
        _worker(0, 1, port, argv)
        return
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, cluster, port, argv))
             for r in range(cluster)]
    for p in procs:
        p.start()
    exit_code = 0
    for p in procs:
        p.join()
        exit_code = exit_code or (p.exitcode or 0)
    sys.exit(exit_code)


if __name__ == "__main__":
    main()

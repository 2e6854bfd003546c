"""Single-node process launcher (reference pssh_start.py:17 analog).

Spawns one python process per GPU with RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*
env (the same contract torchrun provides), plus an optional KV/heartbeat
server in-process.  Multi-node launch uses torch.distributed.run directly;
this module covers the reference's single-node `pssh` path and the elastic
restart loop (pssh_start_elastic.py).
"""
from __future__ import annotations

import os
import subprocess
import sys
import time
from typing import Dict, List, Optional


def launch_local(script: str, nproc: int, master_port: int = 29500,
                 env_extra: Optional[Dict[str, str]] = None,
                 script_args: Optional[List[str]] = None,
                 monitor_restart: bool = False, max_restarts: int = 2):
    """Run `script` as nproc ranks; returns per-rank exit codes.  With
    monitor_restart, dead ranks are restarted up to max_restarts times
    (elastic loop)."""
    procs: List[Optional[subprocess.Popen]] = [None] * nproc
    restarts = [0] * nproc

    def spawn(rank: int):
        env = dict(os.environ)
        env.update(env_extra or {})
        env.update({
            "RANK": str(rank), "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(nproc),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(master_port),
        })
        procs[rank] = subprocess.Popen(
            [sys.executable, script] + (script_args or []), env=env)

    for r in range(nproc):
        spawn(r)
    codes = [None] * nproc
    while any(c is None for c in codes):
        for r, p in enumerate(procs):
            if codes[r] is not None or p is None:
                continue
            rc = p.poll()
            if rc is None:
                continue
            if rc != 0 and monitor_restart and restarts[r] < max_restarts:
                restarts[r] += 1
                spawn(r)
            else:
                codes[r] = rc
        time.sleep(0.2)
    return codes

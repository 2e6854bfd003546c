"""Multi-node SSH launcher (reference python/hetu/rpc/pssh_start.py:17 /
pssh_start_elastic.py).

The reference SSH-spawns one python worker per GPU per host plus a gRPC
DeviceController.  MI355X-native shape: one ssh session per HOST running a
`torch.distributed`-env worker group (one process per local GPU); the
rendezvous master is host 0.  The transport command is injectable
(`ssh_cmd`) so tests can substitute a local shell, and a DEAD host's
workers can be respawned (elastic pool semantics, pssh_workers.py).
"""
from __future__ import annotations

import shlex
import subprocess
import sys
import time
from typing import Dict, List, Optional, Sequence


class Host:
    def __init__(self, addr: str, gpus: int):
        self.addr = addr
        self.gpus = gpus


def _remote_cmd(host: Host, script: str, args: Sequence[str],
                env: Dict[str, str], python: str) -> str:
    exports = " ".join(f"export {k}={shlex.quote(str(v))};"
                       for k, v in env.items())
    argstr = " ".join(shlex.quote(a) for a in args)
    return (f"{exports} {python} -m torch.distributed.run "
            f"--nnodes={env['PSSH_NNODES']} "
            f"--node-rank={env['PSSH_NODE_RANK']} "
            f"--nproc-per-node={host.gpus} "
            f"--master-addr={env['MASTER_ADDR']} "
            f"--master-port={env['MASTER_PORT']} "
            f"{shlex.quote(script)} {argstr}")


class PsshLauncher:
    """Launch `script` across hosts over ssh; poll() monitors the ssh
    sessions; respawn(i) restarts a dead host's worker group."""

    def __init__(self, hosts: List[Host], script: str,
                 script_args: Optional[Sequence[str]] = None,
                 master_port: int = 29500,
                 env_extra: Optional[Dict[str, str]] = None,
                 ssh_cmd: Optional[Sequence[str]] = None,
                 python: str = sys.executable):
        self.hosts = hosts
        self.script = script
        self.args = list(script_args or [])
        self.master_port = master_port
        self.env_extra = dict(env_extra or {})
        # default transport; tests inject ["bash", "-lc"] for local runs
        self.ssh = list(ssh_cmd) if ssh_cmd is not None else \
            ["ssh", "-o", "StrictHostKeyChecking=no"]
        self.python = python
        self.procs: List[Optional[subprocess.Popen]] = [None] * len(hosts)

    def _spawn(self, i: int):
        host = self.hosts[i]
        env = dict(self.env_extra,
                   MASTER_ADDR=self.hosts[0].addr,
                   MASTER_PORT=str(self.master_port),
                   PSSH_NNODES=str(len(self.hosts)),
                   PSSH_NODE_RANK=str(i))
        cmd = _remote_cmd(host, self.script, self.args, env, self.python)
        if self.ssh and self.ssh[0] == "ssh":
            full = self.ssh + [host.addr, cmd]
        else:
            full = self.ssh + [cmd]          # injected local transport
        self.procs[i] = subprocess.Popen(full, stdout=subprocess.PIPE,
                                         stderr=subprocess.STDOUT,
                                         text=True)

    def start(self):
        for i in range(len(self.hosts)):
            self._spawn(i)
        return self

    def poll(self) -> List[Optional[int]]:
        return [p.poll() if p is not None else None for p in self.procs]

    def respawn(self, i: int):
        p = self.procs[i]
        if p is not None and p.poll() is None:
            p.kill()
        self._spawn(i)

    def wait(self, timeout_s: Optional[float] = None) -> List[int]:
        deadline = None if timeout_s is None else time.time() + timeout_s
        codes: List[Optional[int]] = [None] * len(self.procs)
        while any(c is None for c in codes):
            for i, p in enumerate(self.procs):
                if codes[i] is None and p is not None:
                    rc = p.poll()
                    if rc is not None:
                        codes[i] = rc
            if deadline is not None and time.time() > deadline:
                raise TimeoutError(f"pssh wait: {codes}")
            time.sleep(0.2)
        return codes  # type: ignore[return-value]

    def output(self, i: int) -> str:
        p = self.procs[i]
        if p is None or p.stdout is None:
            return ""
        return p.stdout.read()

    def kill(self):
        for p in self.procs:
            if p is not None and p.poll() is None:
                p.kill()

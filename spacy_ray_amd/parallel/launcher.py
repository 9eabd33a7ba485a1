"""Launcher: fork one worker process per GPU and supervise them.

Replaces Ray's actor scheduling: the driver
spawns N ranked processes with the torchrun env contract (RANK / LOCAL_RANK /
WORLD_SIZE / MASTER_ADDR / MASTER_PORT), waits on them, and on any child
failure terminates the rest and exits nonzero (RCCL cannot shrink a
communicator mid-flight — SURVEY.md §5.3 disposition; recovery story is
restart from the last checkpoint)."""
from __future__ import annotations

import os
import socket
import subprocess
import time
from typing import Dict, List, Optional


def find_free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def launch_workers(
    worker_cmd: List[str],
    n_workers: int,
    *,
    master_addr: str = "127.0.0.1",
    master_port: Optional[int] = None,
    nnodes: int = 1,
    node_rank: int = 0,
    env_extra: Optional[Dict[str, str]] = None,
    poll_interval: float = 1.0,
) -> int:
    """Spawn n_workers ranked processes on THIS node; supervise; return exit
    code.  Multi-node: run the same command on every node with the same
    --address (rank-0 node's host:port) and the node's --node-rank; global
    RANK = node_rank * n_workers + local."""
    port = master_port or find_free_port()
    procs: List[subprocess.Popen] = []
    for local in range(n_workers):
        env = dict(os.environ)
        env.update(env_extra or {})
        env.update(
            RANK=str(node_rank * n_workers + local),
            LOCAL_RANK=str(local),
            WORLD_SIZE=str(n_workers * nnodes),
            MASTER_ADDR=master_addr,
            MASTER_PORT=str(port),
        )
        procs.append(subprocess.Popen(worker_cmd, env=env))
    exit_code = 0
    try:
        # poll-until-done (the reference driver's is_running loop,
        # /root/reference/spacy_ray/train_cli.py:88-91, on waitpid instead)
        while procs:
            time.sleep(poll_interval)
            for p in list(procs):
                rc = p.poll()
                if rc is None:
                    continue
                procs.remove(p)
                if rc != 0:
                    exit_code = rc
                    for q in procs:
                        q.terminate()
                    for q in procs:
                        try:
                            q.wait(timeout=20)
                        except subprocess.TimeoutExpired:
                            q.kill()
                    return exit_code
    except KeyboardInterrupt:
        for q in procs:
            q.terminate()
        exit_code = 130
    return exit_code

"""Coordinator — chief re-executes the user script as one worker per GPU.

Reference behavior: autodist/coordinator.py:46-110 (chief re-runs
`python sys.argv` on every worker over SSH with AUTODIST_WORKER /
AUTODIST_STRATEGY_ID env, monitors remote procs in threads, kills all on
failure via os._exit(1)).

MI355X translation: the primary topology is one 8-GPU node, so workers are
LOCAL subprocesses (one per GPU, rank = device index); rendezvous is a TCP
store on 127.0.0.1. Multi-node launch goes through runtime/cluster.py's SSH
exec with the same env protocol.
"""
import atexit
import os
import socket
import subprocess
import sys
import threading
from typing import List

from autodist_amd.const import DEFAULT_MASTER_ADDR
from autodist_amd.utils import logging


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind((DEFAULT_MASTER_ADDR, 0))
        return s.getsockname()[1]


class Coordinator:
    def __init__(self, strategy, resource_spec, cluster=None):
        self.strategy = strategy
        self.resource_spec = resource_spec
        self.cluster = cluster
        self.procs: List[subprocess.Popen] = []
        self._failed = False

    def launch_clients(self, world_size: int, master_port: int):
        """Spawn ranks 1..world_size-1 re-running this script
        (reference launch_clients, coordinator.py:46-90)."""
        base_env = dict(os.environ)
        base_env.update({
            "WORLD_SIZE": str(world_size),
            "MASTER_ADDR": DEFAULT_MASTER_ADDR,
            "MASTER_PORT": str(master_port),
            "AUTODIST_STRATEGY_ID": self.strategy.id,
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        })
        for rank in range(1, world_size):
            env = dict(base_env)
            env["RANK"] = str(rank)
            env["LOCAL_RANK"] = str(rank)
            env["AUTODIST_WORKER"] = DEFAULT_MASTER_ADDR
            cmd = [sys.executable] + sys.argv
            logging.info("launching worker rank %d: %s", rank, " ".join(cmd))
            proc = subprocess.Popen(cmd, env=env, start_new_session=True)
            self.procs.append(proc)
            t = threading.Thread(target=self._proc_wait_async,
                                 args=(proc, rank), daemon=True)
            t.start()
        atexit.register(self.terminate)
        return self

    def _proc_wait_async(self, proc, rank):
        """Kill everything if a worker dies (reference _proc_wait_async,
        coordinator.py:98-110)."""
        ret = proc.wait()
        if ret != 0 and not self._failed:
            self._failed = True
            logging.error("worker rank %d exited with %d — aborting chief",
                          rank, ret)
            self.terminate()
            os._exit(1)

    def join(self):
        """Wait for all workers (reference join, coordinator.py:92-96)."""
        for p in self.procs:
            p.wait()

    def terminate(self):
        for p in self.procs:
            if p.poll() is None:
                try:
                    p.terminate()
                except OSError:
                    pass

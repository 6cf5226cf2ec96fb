"""Cluster — node/process management for multi-node launches.

Reference behavior: autodist/cluster.py:51-374. The reference starts a TF
gRPC server per node (server_starter.py) and drives remote nodes over
paramiko SSH/SFTP. The MI355X translation needs NO per-node graph servers —
the data plane is RCCL over a torch.distributed TCP rendezvous — so Cluster
keeps the reference's responsibilities that still exist:

  * the cluster spec: ordered rank assignment, one rank per GPU, chief node
    hosts the rendezvous (MASTER_ADDR) — replaces cluster.py:54-80's
    deterministic sorted ip->port map,
  * remote exec / file shipping over ssh/scp subprocesses (paramiko is not
    in the image) — cluster.py:316-374,
  * process-group termination on exit — cluster.py:212-216.

AUTODIST_DEBUG_REMOTE makes remote_exec a dry-run (cluster.py:340-341).
"""
import atexit
import os
import subprocess
from typing import Dict, List, Optional

from autodist_amd.const import DEFAULT_MASTER_PORT, ENV
from autodist_amd.resource_spec import ResourceSpec
from autodist_amd.utils import logging
from autodist_amd.utils.network import is_local_address


class Cluster:
    def __init__(self, resource_spec: ResourceSpec):
        self._spec = resource_spec
        self._procs: List[subprocess.Popen] = []
        # deterministic global rank order: nodes sorted, then gpu index
        self.rank_table: List[tuple] = []   # (addr, local_rank, global_rank)
        g = 0
        for addr in resource_spec.nodes:
            for local, _dev in enumerate(resource_spec.node_gpu_devices(addr)
                                         or [None]):
                self.rank_table.append((addr, local, g))
                g += 1
        self.world_size = g
        self.master_addr = resource_spec.chief
        self.master_port = DEFAULT_MASTER_PORT

    @property
    def cluster_spec(self) -> Dict[str, List[str]]:
        """{"worker": ["ip:rank", ...]} (reference cluster_spec shape)."""
        return {"worker": [f"{a}:{r}" for a, _l, r in self.rank_table]}

    def is_chief(self, address: Optional[str] = None) -> bool:
        address = address or self.master_addr
        return address == self._spec.chief

    # -- process launch ----------------------------------------------------
    def start(self, argv: List[str], extra_env: Optional[dict] = None):
        """Launch every non-chief rank: local ranks as subprocesses, remote
        ranks over SSH (reference Cluster.start, cluster.py:160-210)."""
        atexit.register(self.terminate)
        for addr, local_rank, global_rank in self.rank_table:
            if global_rank == 0:
                continue  # chief = this process
            env = {
                "RANK": str(global_rank), "LOCAL_RANK": str(local_rank),
                "WORLD_SIZE": str(self.world_size),
                "MASTER_ADDR": self.master_addr,
                "MASTER_PORT": str(self.master_port),
                "AUTODIST_WORKER": addr,
                "HSA_ENABLE_IPC_MODE_LEGACY": "0",
            }
            if extra_env:
                env.update(extra_env)
            if is_local_address(addr):
                full = dict(os.environ)
                full.update(env)
                proc = subprocess.Popen(argv, env=full,
                                        start_new_session=True)
                self._procs.append(proc)
            else:
                self.remote_exec(addr, argv, env)
        return self

    def terminate(self):
        """SIGTERM all spawned process groups (reference cluster.py:212-216)."""
        for p in self._procs:
            if p.poll() is None:
                try:
                    p.terminate()
                except OSError:
                    pass

    def join(self) -> int:
        rc = 0
        for p in self._procs:
            rc |= p.wait()
        return rc

    # -- remote ops (ssh/scp subprocesses) ---------------------------------
    def _ssh_base(self, addr: str) -> List[str]:
        ssh_conf = self._spec.ssh_config(addr)
        cmd = ["ssh", "-o", "StrictHostKeyChecking=no"]
        if ssh_conf:
            if ssh_conf.key_file:
                cmd += ["-i", ssh_conf.key_file]
            cmd += ["-p", str(ssh_conf.port)]
            target = f"{ssh_conf.username}@{addr}" if ssh_conf.username else addr
        else:
            target = addr
        return cmd + [target]

    def remote_exec(self, addr: str, argv: List[str],
                    env: Optional[dict] = None) -> Optional[subprocess.Popen]:
        """Run a command on a remote node (reference remote_exec,
        cluster.py:316-345)."""
        env = dict(env or {})
        ssh_conf = self._spec.ssh_config(addr)
        if ssh_conf:
            env.update(ssh_conf.shared_envs)
        env_str = " ".join(f"{k}={v}" for k, v in env.items())
        venv = f"source {ssh_conf.python_venv}/bin/activate && " \
            if ssh_conf and ssh_conf.python_venv else ""
        remote_cmd = f"{venv}{env_str} {' '.join(argv)}"
        full = self._ssh_base(addr) + [remote_cmd]
        if ENV.AUTODIST_DEBUG_REMOTE.val:
            logging.info("[dry-run] remote_exec %s: %s", addr, remote_cmd)
            return None
        logging.info("remote_exec %s: %s", addr, remote_cmd)
        proc = subprocess.Popen(full, start_new_session=True)
        self._procs.append(proc)
        return proc

    def remote_copy(self, addr: str, local_path: str, remote_dir: str):
        """Ship a file to a remote node (reference remote_copy,
        cluster.py:360-374)."""
        if ENV.AUTODIST_DEBUG_REMOTE.val:
            logging.info("[dry-run] remote_copy %s -> %s:%s", local_path,
                         addr, remote_dir)
            return
        ssh_conf = self._spec.ssh_config(addr)
        cmd = ["scp", "-o", "StrictHostKeyChecking=no"]
        if ssh_conf and ssh_conf.key_file:
            cmd += ["-i", ssh_conf.key_file]
        if ssh_conf:
            cmd += ["-P", str(ssh_conf.port)]
        target = f"{ssh_conf.username}@{addr}" if ssh_conf and \
            ssh_conf.username else addr
        self._ssh_run([*self._ssh_base(addr), f"mkdir -p {remote_dir}"])
        subprocess.run(cmd + [local_path, f"{target}:{remote_dir}/"],
                       check=True)

    @staticmethod
    def _ssh_run(cmd: List[str]):
        subprocess.run(cmd, check=False)


class SSHCluster(Cluster):
    """Name kept for reference parity (cluster.py:271-374)."""

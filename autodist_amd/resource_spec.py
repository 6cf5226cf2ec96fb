"""Cluster resource model: YAML spec -> DeviceSpec graph.

Reference behavior: autodist/resource_spec.py:45-331 (nodes with address/gpus/
cpus/chief/ssh_config/network_bandwidth; DeviceSpec strings "ip:GPU:k").

MI355X-native notes: the primary topology is ONE node with 8 MI355X GPUs on an
xGMI mesh (7 p2p links/GPU, ~153 GB/s each, ~8 TB/s HBM3E per GPU, 288 GB).
When no YAML is given we synthesize a spec from the local machine
(torch.cuda.device_count()), which is the idiomatic single-node launch.
SSH config is parsed and retained for multi-node launches.
"""
import enum
import os
from typing import Dict, List, Optional

import yaml

# xGMI defaults for the cost model (GB/s); overridable per-node in YAML.
XGMI_LINK_GBPS = 153.0
XGMI_LINKS_PER_GPU = 7
DEFAULT_NETWORK_BANDWIDTH_GBPS = 1.0  # reference default: 1 GbE (resource_spec.py:209-215)
HBM_GBPS = 6300.0  # achievable HBM3E bandwidth per MI355X


class DeviceType(enum.Enum):
    CPU = "CPU"
    GPU = "GPU"


class DeviceSpec:
    """One device: "<host>:<type>:<index>" (reference resource_spec.py:218-277)."""

    def __init__(self, host: str, device_type: DeviceType = DeviceType.GPU,
                 device_index: int = 0):
        self.host = host
        self.device_type = device_type
        self.device_index = int(device_index)

    def name_string(self) -> str:
        return f"{self.host}:{self.device_type.value}:{self.device_index}"

    @classmethod
    def from_string(cls, s: str) -> "DeviceSpec":
        parts = s.split(":")
        if len(parts) == 1:
            return cls(parts[0], DeviceType.CPU, 0)
        if len(parts) == 2:  # "host:index" => GPU
            return cls(parts[0], DeviceType.GPU, int(parts[1]))
        return cls(parts[0], DeviceType(parts[1].upper()), int(parts[2]))

    def __eq__(self, other):
        return isinstance(other, DeviceSpec) and self.name_string() == other.name_string()

    def __hash__(self):
        return hash(self.name_string())

    def __repr__(self):
        return f"DeviceSpec({self.name_string()})"


class SSHConfig:
    """SSH group config (reference resource_spec.py:280-331)."""

    def __init__(self, info: dict):
        self.username = info.get("username", "")
        self.port = info.get("port", 22)
        self.key_file = info.get("key_file")
        self.python_venv = info.get("python_venv", "")
        self.shared_envs = dict(info.get("shared_envs", {}))
        # Propagate the IPC mode flag required for RCCL dmabuf IPC on this pool.
        self.shared_envs.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")


class ResourceSpec:
    """Parse a resource-spec YAML into node/device/bandwidth maps."""

    def __init__(self, resource_file: Optional[str] = None):
        self._nodes: Dict[str, dict] = {}
        self._devices: Dict[str, DeviceSpec] = {}
        self._chief: Optional[str] = None
        self._ssh_configs: Dict[str, SSHConfig] = {}
        self._bandwidth: Dict[str, float] = {}   # GB/s network per node
        self._xgmi_link_gbps: float = XGMI_LINK_GBPS
        self._xgmi_links: int = XGMI_LINKS_PER_GPU
        if resource_file:
            if not os.path.exists(resource_file):
                raise FileNotFoundError(f"resource spec not found: {resource_file}")
            with open(resource_file, "r", encoding="utf-8") as f:
                self._from_dict(yaml.safe_load(f) or {})
        else:
            self._from_local()

    # -- construction ------------------------------------------------------
    def _from_local(self):
        """Synthesize a single-node spec from the local machine."""
        try:
            import torch
            n_gpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
        except Exception:  # pragma: no cover
            n_gpu = 0
        node = {"address": "127.0.0.1", "gpus": list(range(n_gpu)),
                "cpus": [0], "chief": True}
        self._add_node(node)

    def _from_dict(self, d: dict):
        nodes = d.get("nodes", [])
        for i, node in enumerate(nodes):
            if "chief" not in node and i == 0 and len(nodes) == 1:
                node["chief"] = True
            self._add_node(node)
        for name, info in (d.get("ssh", {}) or {}).items():
            self._ssh_configs[name] = SSHConfig(info or {})
        gpu_info = d.get("gpu", {}) or {}
        self._xgmi_link_gbps = float(gpu_info.get("xgmi_link_gbps", XGMI_LINK_GBPS))
        self._xgmi_links = int(gpu_info.get("xgmi_links_per_gpu", XGMI_LINKS_PER_GPU))
        if self._chief is None:
            raise ValueError("resource spec must mark exactly one node as chief")

    def _add_node(self, node: dict):
        addr = str(node["address"])
        if addr in self._nodes:
            raise ValueError(f"duplicate node address {addr}")
        gpus = node.get("gpus", [])
        if isinstance(gpus, int):
            gpus = list(range(gpus))
        cpus = node.get("cpus", [0])
        if isinstance(cpus, int):
            cpus = list(range(cpus))
        self._nodes[addr] = {
            "address": addr,
            "gpus": list(gpus),
            "cpus": list(cpus),
            "chief": bool(node.get("chief", False)),
            "ssh_config": node.get("ssh_config"),
        }
        if node.get("chief"):
            if self._chief is not None and self._chief != addr:
                raise ValueError("only one chief allowed")
            self._chief = addr
        for idx in gpus:
            dev = DeviceSpec(addr, DeviceType.GPU, idx)
            self._devices[dev.name_string()] = dev
        for idx in cpus:
            dev = DeviceSpec(addr, DeviceType.CPU, idx)
            self._devices[dev.name_string()] = dev
        self._bandwidth[addr] = float(
            node.get("network_bandwidth", DEFAULT_NETWORK_BANDWIDTH_GBPS))

    # -- queries -----------------------------------------------------------
    @property
    def chief(self) -> str:
        return self._chief or next(iter(self._nodes), "127.0.0.1")

    @property
    def nodes(self) -> List[str]:
        return sorted(self._nodes.keys())

    @property
    def num_nodes(self) -> int:
        return len(self._nodes)

    @property
    def gpu_devices(self):
        """Sorted (name, DeviceSpec) pairs of all GPUs."""
        return sorted(((k, v) for k, v in self._devices.items()
                       if v.device_type == DeviceType.GPU))

    @property
    def cpu_devices(self):
        return sorted(((k, v) for k, v in self._devices.items()
                       if v.device_type == DeviceType.CPU))

    @property
    def num_gpus(self) -> int:
        return len([1 for _, v in self._devices.items()
                    if v.device_type == DeviceType.GPU])

    def node_gpu_devices(self, addr: str):
        return [DeviceSpec(addr, DeviceType.GPU, i)
                for i in self._nodes[addr]["gpus"]]

    def node_cpu_device(self, addr: str) -> DeviceSpec:
        return DeviceSpec(addr, DeviceType.CPU, self._nodes[addr]["cpus"][0])

    def network_bandwidth(self, addr: str) -> float:
        return self._bandwidth.get(addr, DEFAULT_NETWORK_BANDWIDTH_GBPS)

    @property
    def xgmi_link_gbps(self) -> float:
        return self._xgmi_link_gbps

    @property
    def xgmi_links_per_gpu(self) -> int:
        return self._xgmi_links

    def ssh_config(self, node_addr: str) -> Optional[SSHConfig]:
        key = self._nodes.get(node_addr, {}).get("ssh_config")
        return self._ssh_configs.get(key) if key else None

    @property
    def ssh_group(self) -> Dict[str, SSHConfig]:
        return dict(self._ssh_configs)

    def __repr__(self):
        return (f"ResourceSpec(nodes={self.nodes}, gpus={self.num_gpus}, "
                f"chief={self.chief})")

"""ResourceSpec / DeviceSpec parsing tests (reference tests/test_resource_spec.py,
test_device_spec.py)."""
import pytest

from autodist_amd.resource_spec import DeviceSpec, DeviceType, ResourceSpec


def test_parse_single_node(tmp_gpu_resource_spec):
    rs = ResourceSpec(tmp_gpu_resource_spec)
    assert rs.num_nodes == 1
    assert rs.num_gpus == 8
    assert rs.chief == "127.0.0.1"
    names = [k for k, _ in rs.gpu_devices]
    assert names[0] == "127.0.0.1:GPU:0"
    assert len(names) == 8


def test_parse_multi_node(tmp_path):
    p = tmp_path / "r.yml"
    p.write_text(
        "nodes:\n"
        "  - address: 10.0.0.1\n    gpus: [0, 1]\n    chief: true\n"
        "    network_bandwidth: 12.5\n"
        "  - address: 10.0.0.2\n    gpus: [0, 1]\n    ssh_config: conf\n"
        "ssh:\n  conf:\n    username: root\n    key_file: /k\n    port: 22\n")
    rs = ResourceSpec(str(p))
    assert rs.num_nodes == 2
    assert rs.num_gpus == 4
    assert rs.chief == "10.0.0.1"
    assert rs.network_bandwidth("10.0.0.1") == 12.5
    ssh = rs.ssh_config("10.0.0.2")
    assert ssh.username == "root"
    assert ssh.shared_envs["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"
    assert rs.ssh_config("10.0.0.1") is None


def test_chief_required(tmp_path):
    p = tmp_path / "r.yml"
    p.write_text("nodes:\n  - address: a\n  - address: b\n")
    with pytest.raises(ValueError):
        ResourceSpec(str(p))


def test_duplicate_node_rejected(tmp_path):
    p = tmp_path / "r.yml"
    p.write_text("nodes:\n  - address: a\n    chief: true\n  - address: a\n")
    with pytest.raises(ValueError):
        ResourceSpec(str(p))


def test_device_spec_string_roundtrip():
    d = DeviceSpec.from_string("127.0.0.1:GPU:3")
    assert d.device_type == DeviceType.GPU
    assert d.device_index == 3
    assert d.name_string() == "127.0.0.1:GPU:3"
    c = DeviceSpec.from_string("hostonly")
    assert c.device_type == DeviceType.CPU


def test_local_synthesized_spec():
    rs = ResourceSpec()  # no yaml: local machine (CPU-only in CI)
    assert rs.num_nodes == 1

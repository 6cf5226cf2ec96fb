"""Deterministic collective keys.

Reference behavior: autodist/kernel/synchronization/collective_key.py:26-70
(singleton mapping device-set -> group_key and var name -> instance_key via
md5 % INT32_MAX). RCCL needs every rank to enqueue collectives in an
identical order; deterministic keys give buckets/synchronizers a canonical
sort order independent of hook firing time.
"""
import hashlib
import threading

from autodist_amd.const import MAX_INT32

_collective_keys = None
_lock = threading.Lock()


class CollectiveKey:
    def __init__(self, group_leader_key: int = 0):
        self._group_key = group_leader_key
        self._group_map = {}
        self._instance_map = {}

    def generate_group_key(self, devices) -> int:
        """Device-set -> incrementing group key (reference collective_key.py:52-58)."""
        key = ",".join(sorted(devices))
        if key not in self._group_map:
            self._group_key += 1
            self._group_map[key] = self._group_key
        return self._group_map[key]

    def generate_instance_key(self, var_name: str) -> int:
        """Var name -> md5-derived stable instance key (reference
        collective_key.py:60-70)."""
        if var_name not in self._instance_map:
            digest = hashlib.md5(var_name.encode()).hexdigest()
            self._instance_map[var_name] = int(digest, 16) % MAX_INT32
        return self._instance_map[var_name]


def get_collective_keys() -> CollectiveKey:
    global _collective_keys
    if _collective_keys is None:
        with _lock:
            if _collective_keys is None:
                _collective_keys = CollectiveKey()
    return _collective_keys

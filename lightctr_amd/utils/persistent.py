"""Persistence substrates.

  * PersistentBuffer — file-backed mmap write/read buffer (parity with
    /root/reference/LightCTR/common/persistent_buffer.h); used as the
    durable staging area for checkpoint shards.
  * ShmHashTable — fixed-slot multi-hash table in shared memory (parity
    with /root/reference/LightCTR/util/shm_hashtable.h: k prime-modulus
    sub-tables, CAS-free slot claim via atomic-ish write-then-verify on a
    memory-mapped file usable across processes on one host).
"""

from __future__ import annotations

import mmap
import os
import struct

import numpy as np


class PersistentBuffer:
    """Append/read buffer backed by a memory-mapped file."""

    def __init__(self, path: str, capacity: int = 1 << 20):
        self.path = path
        exists = os.path.exists(path)
        self._f = open(path, "r+b" if exists else "w+b")
        if not exists or os.path.getsize(path) < capacity + 8:
            self._f.truncate(capacity + 8)
        self._mm = mmap.mmap(self._f.fileno(), 0)
        self.capacity = len(self._mm) - 8
        if not exists:
            self._set_size(0)

    def _set_size(self, n: int):
        self._mm[:8] = struct.pack("<q", n)

    @property
    def size(self) -> int:
        return struct.unpack("<q", self._mm[:8])[0]

    def write(self, data: bytes) -> None:
        n = self.size
        assert n + len(data) <= self.capacity, "persistent buffer full"
        self._mm[8 + n:8 + n + len(data)] = data
        self._set_size(n + len(data))

    def read_all(self) -> bytes:
        return bytes(self._mm[8:8 + self.size])

    def clear(self) -> None:
        self._set_size(0)

    def flush(self) -> None:
        self._mm.flush()

    def close(self) -> None:
        self._mm.flush()
        self._mm.close()
        self._f.close()


_PRIMES = [1000003, 999983, 999979, 999961]


class ShmHashTable:
    """uint64 -> float32[value_dim] table over k prime-modulus sub-tables
    in a shared mmap file (multi-process safe for the single-writer /
    multi-reader pattern the reference uses)."""

    def __init__(self, path: str, slots_per_table: int = 4096,
                 n_tables: int = 4, value_dim: int = 1):
        self.n_tables = n_tables
        self.slots = slots_per_table
        self.value_dim = value_dim
        self.primes = _PRIMES[:n_tables]
        entry = 8 + 4 * value_dim  # key + values
        total = 16 + n_tables * slots_per_table * entry
        exists = os.path.exists(path)
        self._f = open(path, "r+b" if exists else "w+b")
        if not exists or os.path.getsize(path) < total:
            self._f.truncate(total)
        self._mm = mmap.mmap(self._f.fileno(), 0)
        self._keys = np.frombuffer(
            self._mm, dtype=np.uint64, offset=16,
            count=n_tables * slots_per_table * (entry // 8)
        ).reshape(n_tables * slots_per_table, entry // 8)[:, 0:1]
        # simpler layout: keys array then values array
        self._mm_keys = np.frombuffer(
            self._mm, dtype=np.uint64, offset=16,
            count=n_tables * slots_per_table)
        voff = 16 + n_tables * slots_per_table * 8
        if len(self._mm) < voff + n_tables * slots_per_table * 4 * value_dim:
            self._f.truncate(voff + n_tables * slots_per_table * 4
                             * value_dim)
            self._mm = mmap.mmap(self._f.fileno(), 0)
            self._mm_keys = np.frombuffer(
                self._mm, dtype=np.uint64, offset=16,
                count=n_tables * slots_per_table)
        self._mm_vals = np.frombuffer(
            self._mm, dtype=np.float32, offset=voff,
            count=n_tables * slots_per_table * value_dim
        ).reshape(n_tables * slots_per_table, value_dim)

    def _slot(self, key: int, t: int) -> int:
        return t * self.slots + (key * 2654435761 % self.primes[t]) \
            % self.slots

    def put(self, key: int, value) -> bool:
        # 0 is the empty-slot marker; shift by 1 (injective) instead of
        # OR-ing bit 0, which aliased every even key k with k+1.
        key = int(key) + 1
        assert 0 < key <= 0xFFFFFFFFFFFFFFFF, "key out of u64 range"
        for t in range(self.n_tables):
            s = self._slot(key, t)
            k = int(self._mm_keys[s])
            if k == 0 or k == key:
                # numpy view is read-only from mmap buffer in some paths;
                # write through the mmap directly
                off = 16 + s * 8
                self._mm[off:off + 8] = struct.pack("<Q", key)
                voff = 16 + len(self._mm_keys) * 8 + s * 4 * self.value_dim
                self._mm[voff:voff + 4 * self.value_dim] = np.asarray(
                    value, dtype=np.float32).tobytes()
                return True
        return False  # all sub-tables collided

    def get(self, key: int):
        key = int(key) + 1
        for t in range(self.n_tables):
            s = self._slot(key, t)
            if int(self._mm_keys[s]) == key:
                return self._mm_vals[s].copy()
        return None

    def close(self):
        self._mm.flush()
        self._f.close()

"""Tensor naming and key assignment.

Reproduces the reference's naming / key-space layout:

- canonical tensor names ``byteps.Gradient.<param>``, ``byteps.Parameter.<n>``,
  ``byteps.AsyncParam.<n>`` (reference torch/ops.cc:37-43,104);
- declared keys assigned in sorted-declaration order so all ranks agree
  without communication (reference common/global.cc:412-429);
- partition keys ``declared_key << 16 | part`` — up to 2^16 partitions per
  tensor (reference common/operations.cc:306-311);
- server assignment by djb2 hash of the partition key with per-server load
  accounting (reference common/global.cc:628-677).
"""

from __future__ import annotations

from typing import Dict, List, Optional

PART_BITS = 16
MAX_PARTS = 1 << PART_BITS


def gradient_name(param_name: str) -> str:
    return "byteps.Gradient." + param_name


def parameter_name(name: str) -> str:
    return "byteps.Parameter." + name


def async_param_name(name: str) -> str:
    return "byteps.AsyncParam." + name


def partition_key(declared_key: int, part: int) -> int:
    assert 0 <= part < MAX_PARTS
    return (declared_key << PART_BITS) | part


def declared_key_of(pkey: int) -> int:
    return pkey >> PART_BITS


def part_of(pkey: int) -> int:
    return pkey & (MAX_PARTS - 1)


def djb2(key: int) -> int:
    """djb2 over the decimal string of the key — the reference's default
    server-assignment hash (reference common/global.cc:628-677)."""
    h = 5381
    for ch in str(key):
        h = ((h * 33) + ord(ch)) & 0xFFFFFFFFFFFFFFFF
    return h


def sdbm(key: int) -> int:
    h = 0
    for ch in str(key):
        h = (ord(ch) + (h << 6) + (h << 16) - h) & 0xFFFFFFFFFFFFFFFF
    return h


def _naive(key: int) -> int:
    return key


_HASH_FNS = {"djb2": djb2, "sdbm": sdbm, "naive": _naive,
             "built_in": lambda k: hash(str(k)) & 0xFFFFFFFFFFFFFFFF}


def key_hash(key: int, fn: str = "") -> int:
    """Pluggable key→server hash (reference BYTEPS_KEY_HASH_FN,
    common/global.cc:628-677: naive/built-in/djb2/sdbm).  ``built_in``
    is process-local (PYTHONHASHSEED-dependent) — use djb2/sdbm across
    machines."""
    import os
    name = fn or os.environ.get(
        "BPS_KEY_HASH_FN", os.environ.get("BYTEPS_KEY_HASH_FN", "djb2"))
    return _HASH_FNS.get(name, djb2)(key)


class NameRegistry:
    """Deterministic name → declared-key table.

    Keys are assigned incrementally in declaration-arrival order
    (reference common/global.cc:412-429); callers must declare in the same
    deterministic order on every worker (the reference's Python plugins
    declare in sorted passes for exactly this reason,
    torch/__init__.py:95-100 — our engine declares in model registration
    order, identical across ranks).  Once assigned, a key never changes —
    elastic resume re-declares in original order to keep keys stable
    (reference common/global.cc:431-436).
    """

    def __init__(self) -> None:
        self._keys: Dict[str, int] = {}
        self._declare_order: List[str] = []

    def declare(self, name: str) -> int:
        if name not in self._keys:
            self._keys[name] = len(self._declare_order)
            self._declare_order.append(name)
        return self._keys[name]

    def key(self, name: str) -> int:
        return self._keys[name]

    def __contains__(self, name: str) -> bool:
        return name in self._keys

    def __len__(self) -> int:
        return len(self._keys)

    @property
    def declare_order(self) -> List[str]:
        return list(self._declare_order)

    def redeclare_all(self) -> None:
        """Elastic resume: re-declare every known tensor in original order
        (reference common/operations.cc:96-119)."""
        order = self._declare_order
        self._keys, self._declare_order = {}, []
        for n in order:
            self.declare(n)


class ServerAssigner:
    """Key → server with hash + greedy weighted load balancing.

    The reference hashes first and falls back to accumulated-load
    accounting (common/global.cc:660-667); mixed mode gives colocated
    servers (those sharing a machine with a worker) a smaller share
    (common/global.cc:566-596).  Here: each server has a capacity weight;
    a key goes to its hash-preferred server unless that server's
    *effective* load (bytes/weight) exceeds the least-loaded one by more
    than the partition size — deterministic given identical declaration
    sequences on all workers.

    Mixed mode: ``BPS_ENABLE_MIXED_MODE=1`` +
    ``BPS_COLOCATED_SERVERS=i,j,...`` (indices into the server list);
    colocated servers get weight ``BPS_MIXED_RATIO`` (default 0.5).
    """

    def __init__(self, num_servers: int,
                 weights: Optional[List[float]] = None) -> None:
        import os
        self.num_servers = max(1, num_servers)
        if weights is None:
            weights = [1.0] * self.num_servers
            if os.environ.get("BPS_ENABLE_MIXED_MODE",
                              os.environ.get("BYTEPS_ENABLE_MIXED_MODE",
                                             "0")) == "1":
                ratio = float(os.environ.get("BPS_MIXED_RATIO", "0.5"))
                colocated = os.environ.get("BPS_COLOCATED_SERVERS", "")
                for tok in colocated.split(","):
                    tok = tok.strip()
                    if tok.isdigit() and int(tok) < self.num_servers:
                        weights[int(tok)] = ratio
        self.weights = weights
        self.load = [0.0] * self.num_servers
        self.table: Dict[int, int] = {}

    def _eff(self, s: int) -> float:
        return self.load[s] / max(self.weights[s], 1e-9)

    def assign(self, pkey: int, nbytes: int) -> int:
        if pkey in self.table:
            return self.table[pkey]
        pref = key_hash(pkey) % self.num_servers
        least = min(range(self.num_servers), key=self._eff)
        srv = pref if self._eff(pref) - self._eff(least) <= \
            nbytes / max(self.weights[pref], 1e-9) else least
        self.table[pkey] = srv
        self.load[srv] += nbytes
        return srv

"""Tensor partition planning.

The reference splits each tensor into fixed-size parts pipelined
independently (reference common/operations.cc:140-180).  Here partitioning
serves two roles:

- **bucketing**: many small parameters are packed into one flat bucket
  buffer (the unit of collective ops / PS push-pull);
- **splitting**: tensors larger than the partition size are split across
  several partitions so push of part 0 overlaps reduce of part 1.

Partition sizes are element counts aligned so that every partition is
divisible by the intra-node world size (reduce-scatter shards stay equal
and aligned; the reference instead special-cased a ``left_elem`` remainder,
common/core_loops.cc:216-217 — alignment removes that branch entirely).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Sequence, Tuple


@dataclass(frozen=True)
class Span:
    """A parameter's slice inside a partition's flat buffer."""
    param_index: int         # index into the registration-order param list
    offset: int              # element offset inside the partition buffer
    numel: int


@dataclass
class PartitionPlan:
    """One schedulable unit: a flat buffer of `numel` elements holding one
    or more parameter spans.  `priority` — higher runs first."""
    index: int
    numel: int
    spans: List[Span]
    priority: int = 0


def _align_up(x: int, a: int) -> int:
    return (x + a - 1) // a * a


def plan_partitions(
    sizes: Sequence[int],
    partition_elems: int,
    align: int = 64,
) -> List[PartitionPlan]:
    """Pack parameters (given in *bucketing order* — callers pass reverse
    registration order so the first gradients produced by backward complete
    first) into partitions of at most ``partition_elems`` elements.

    A parameter larger than ``partition_elems`` is split across consecutive
    partitions.  Each span offset is aligned to ``align`` elements so HIP
    vectorized (`short8`) kernels never straddle partition buffers
    unaligned.  Priorities are assigned in pack order: partition 0 gets the
    highest priority (it holds the gradients backward produces first).
    """
    assert partition_elems >= align
    plans: List[PartitionPlan] = []
    cur_spans: List[Span] = []
    cur_off = 0

    def flush() -> None:
        nonlocal cur_spans, cur_off
        if cur_spans:
            plans.append(PartitionPlan(len(plans), cur_off, cur_spans))
            cur_spans, cur_off = [], 0

    for pidx, size in enumerate(sizes):
        remaining = size
        while remaining > 0:
            space = partition_elems - cur_off
            if space < align:
                flush()
                space = partition_elems
            take = min(remaining, space)
            cur_spans.append(Span(pidx, cur_off, take))
            cur_off = _align_up(cur_off + take, align)
            remaining -= take
    flush()

    n = len(plans)
    for p in plans:
        p.priority = n - p.index   # earlier partition = higher priority
        # round buffer up so reduce-scatter shards divide evenly later
        p.numel = _align_up(p.numel, align)
    return plans


def shard_range(numel: int, world: int, rank: int) -> Tuple[int, int]:
    """[begin, end) of ``rank``'s reduce-scatter shard of a buffer whose
    length is already aligned to ``world`` (plan_partitions guarantees
    align=64 ≥ any world size we target; callers must pass align ≥ world)."""
    assert numel % world == 0, (numel, world)
    per = numel // world
    return rank * per, (rank + 1) * per

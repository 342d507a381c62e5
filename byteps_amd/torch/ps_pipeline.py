"""Hierarchical PS pipeline: reduce-scatter → push/pull KV → all-gather.

Placeholder until the C++ KV transport lands (see DESIGN.md build order);
PS mode is activated by ``BPS_NUM_SERVER>0`` or ``BPS_FORCE_DISTRIBUTED``.
"""

from __future__ import annotations


def get_pipeline(engine):
    raise NotImplementedError(
        "PS pipeline is not built yet — unset BPS_FORCE_DISTRIBUTED / "
        "BPS_NUM_SERVER to use the RCCL-only path")


def get_tensor_pipeline():
    raise NotImplementedError(
        "PS pipeline is not built yet — unset BPS_FORCE_DISTRIBUTED / "
        "BPS_NUM_SERVER to use the RCCL-only path")

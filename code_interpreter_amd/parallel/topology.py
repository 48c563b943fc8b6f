"""MI355X node topology model and collective sizing.

The 8-GPU MI355X node is FULLY CONNECTED over xGMI: each GPU has 7
point-to-point links of ~153 GB/s (no switch). That shapes collective
tuning differently from an NVSwitch node:

- a single-ring all-reduce is bounded by ONE link's bandwidth per GPU
  (the ring uses 1 of the 7 links each direction), so the achievable
  "bus bandwidth" of a ring is ~153 GB/s regardless of world size;
- RCCL recovers more of the aggregate by running multiple rings/channels
  across the other links; what the launcher controls is BUCKET SIZE:
  buckets must be large enough to amortize per-collective launch/latency
  cost, and small enough that several are in flight to overlap with
  compute (e.g. backward) and to pipeline across channels.

These helpers centralize that arithmetic for mgpu (sandbox multi-GPU
jobs) and for tests; everything is a plain model with the measured
constants in one place, overridable by env for other node types.
"""

import os

# per-link unidirectional xGMI bandwidth, GB/s (MI355X: 7 links/GPU)
XGMI_LINK_GBPS = float(os.environ.get("APP_XGMI_LINK_GBPS", 153.0))
XGMI_LINKS_PER_GPU = int(os.environ.get("APP_XGMI_LINKS_PER_GPU", 7))

# empirically sensible flight depth: >= 4 buckets in flight keeps the
# links busy while compute proceeds; single huge buckets serialize
MIN_BUCKETS_IN_FLIGHT = 4
DEFAULT_BUCKET_BYTES = 64 << 20
MIN_BUCKET_BYTES = 4 << 20


def gpu_inventory():
    """Visible GPUs as [(index, name, total_mem_bytes)]; empty without
    torch or devices (CPU boxes, unit tests)."""
    try:
        import torch
    except ImportError:
        return []
    if not torch.cuda.is_available():
        return []
    out = []
    for i in range(torch.cuda.device_count()):
        props = torch.cuda.get_device_properties(i)
        out.append((i, props.name, props.total_memory))
    return out


def ring_allreduce_seconds(nbytes: int, world_size: int,
                           link_gbps: float = XGMI_LINK_GBPS) -> float:
    """Time model for one ring all-reduce of `nbytes` per rank: each GPU
    sends 2*(N-1)/N of the payload over its ring link."""
    if world_size <= 1:
        return 0.0
    wire = 2.0 * (world_size - 1) / world_size * nbytes
    return wire / (link_gbps * 1e9)


def recommended_bucket_bytes(total_bytes: int,
                             world_size: int = 8) -> int:
    """Bucket size for a bucketed all-reduce of `total_bytes`: the 64 MB
    default, shrunk so at least MIN_BUCKETS_IN_FLIGHT buckets exist
    (small payloads should still pipeline), floored at MIN_BUCKET_BYTES
    (smaller buckets are launch-latency-bound on xGMI)."""
    if world_size <= 1 or total_bytes <= 0:
        return DEFAULT_BUCKET_BYTES
    cap = max(MIN_BUCKET_BYTES, total_bytes // MIN_BUCKETS_IN_FLIGHT)
    return min(DEFAULT_BUCKET_BYTES, cap)

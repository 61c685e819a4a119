"""Analytic performance models for MI355X (capability parity with
Triton-distributed kernels/amd/perf_model.py:30-101 NUM_XCDS/TFLOPS
estimators and kernels/nvidia/comm_perf_model.py:94-112 link-bandwidth
method selection — re-derived for CDNA4/xGMI).

Measured constants come from this repo's own GPU runs (profiles/README.md).
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class MI355X:
    num_xcds: int = 8
    num_cus: int = 256
    lds_per_cu_kb: int = 160
    l2_per_xcd_mb: int = 4
    llc_mb: int = 256
    hbm_gb: int = 288
    hbm_peak_tbps: float = 8.0
    hbm_achievable_tbps: float = 6.3     # measured float4-copy ceiling
    bf16_dense_peak_tf: float = 2495.0   # dense (AMD spec halves w/o 2:1)
    fp8_dense_peak_tf: float = 5000.0
    xgmi_links: int = 7                  # point-to-point, per GPU
    xgmi_link_gbps: float = 153.0        # per direction per link
    max_clock_ghz: float = 2.4


ARCH = MI355X()

# measured efficiency of our kernel tiers (profiles/README.md ladder)
GEMM_EFF = {
    "256": 0.42,     # 256^2 K-slice ring at large shapes (~1000-1180 TF)
    "128": 0.23,     # v1 fallback
    "splitk": 0.16,  # occupancy-starved decode shapes
}


def gemm_time_us(m: int, n: int, k: int, tier: str = "256") -> float:
    """Max of the compute roofline at the tier's measured efficiency and
    the staging-traffic roofline (A+B read once through cache/HBM)."""
    flops = 2.0 * m * n * k
    t_compute = flops / (ARCH.bf16_dense_peak_tf * 1e12 * GEMM_EFF[tier])
    bytes_min = 2.0 * (m * k + n * k + m * n)
    t_mem = bytes_min / (ARCH.hbm_achievable_tbps * 1e12)
    return max(t_compute, t_mem) * 1e6


def ag_push_time_us(shard_bytes: int, world: int) -> float:
    """Full-mesh push: each rank streams its shard to world-1 peers on
    independent xGMI links concurrently -> time = shard / link_bw (links
    run in parallel; SDMA saturates a link per peer)."""
    if world <= 1:
        return 0.0
    return shard_bytes / (ARCH.xgmi_link_gbps * 1e9) * 1e6


def allreduce_time_us(nbytes: int, world: int, method: str = "auto") -> float:
    """one_shot: every rank pushes full buffer to all peers + local reduce.
    two_shot: scatter slices + reduced-slice broadcast (2x(W-1)/W per link).
    """
    if world <= 1:
        return 0.0
    link = ARCH.xgmi_link_gbps * 1e9
    t_one = nbytes / link + (world * nbytes) / (ARCH.hbm_achievable_tbps
                                                * 1e12)
    slice_b = nbytes / world
    t_two = 2 * slice_b / link * (world - 1) / max(world - 1, 1) \
        + 2 * nbytes / (ARCH.hbm_achievable_tbps * 1e12)
    if method == "one_shot":
        return t_one * 1e6
    if method == "two_shot":
        return t_two * 1e6
    return min(t_one, t_two) * 1e6


def choose_ar_method(nbytes: int, world: int) -> str:
    return "one_shot" if allreduce_time_us(nbytes, world, "one_shot") <= \
        allreduce_time_us(nbytes, world, "two_shot") else "two_shot"


# measured on-device protocol latencies (profiles/README.md; dev-box runs)
XGMI_ONEWAY_LATENCY_US = 1.6     # small-put visible-at-peer latency
KERNEL_LAUNCH_US = 2.0           # eager launch overhead (graphs: ~0)


def ll_allgather_time_us(payload_bytes: int, world: int) -> float:
    """Flag-in-payload allgather: one crossing carries data+signal, so
    latency ~ max(link transfer of 2x payload, one xGMI latency). The 2x
    is the (data, tag) interleave."""
    if world <= 1:
        return KERNEL_LAUNCH_US
    wire = 2.0 * payload_bytes / (ARCH.xgmi_link_gbps * 1e9) * 1e6
    return max(wire, XGMI_ONEWAY_LATENCY_US) + KERNEL_LAUNCH_US


def choose_ag_method(payload_bytes: int, world: int) -> str:
    """'ll' (flag-in-payload, 2x wire bytes, zero signal round-trips) for
    small payloads; 'push' (SDMA bulk + flag copy) once the doubled wire
    bytes cost more than the saved signal latency."""
    if world <= 1:
        return "push"
    t_ll = ll_allgather_time_us(payload_bytes, world)
    t_push = (ag_push_time_us(payload_bytes, world)
              + XGMI_ONEWAY_LATENCY_US + 2 * KERNEL_LAUNCH_US)
    return "ll" if t_ll < t_push else "push"


def reduce_scatter_time_us(nbytes: int, world: int) -> float:
    """Push-your-segments ((world-1)/world of the buffer leaves over
    world-1 parallel links) + local reduce (world segments read, one
    written)."""
    if world <= 1:
        return KERNEL_LAUNCH_US
    seg = nbytes / world
    t_push = seg / (ARCH.xgmi_link_gbps * 1e9) * 1e6
    t_reduce = (nbytes + seg) / (ARCH.hbm_achievable_tbps * 1e12) * 1e6
    return t_push + t_reduce + XGMI_ONEWAY_LATENCY_US

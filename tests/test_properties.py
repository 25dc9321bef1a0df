"""Property-based tests (hypothesis) for load-bearing invariants:
the LDS swizzle bijection the attention kernels rely on, k8s quantity
parsing, exit-code classification totality, and CRD round-trips."""
from __future__ import annotations

from hypothesis import given, settings, strategies as st

from torch_on_k8s_amd.controlplane import failover as fo
from torch_on_k8s_amd.controlplane.jobspec import parse_quantity


def swz_off(row: int, row_bytes: int, byte_in_row: int) -> int:
    """Mirror of ops/csrc/attention.hip swz_off (full-offset XOR)."""
    return (row * row_bytes + byte_in_row) ^ ((row & 15) << 4)


def test_swizzle_bijection_rowmajor_tiles():
    """The XOR swizzle must be a bijection INSIDE the tile for every
    layout the kernels stage (row-major 256B rows x 64, transposed
    128B rows x 128) — a collision would silently corrupt LDS tiles
    (the r1 'swizzle overflow' bug class)."""
    for rows, row_bytes in ((64, 256), (128, 128), (64, 128), (128, 256)):
        seen = set()
        for r in range(rows):
            for slot in range(0, row_bytes, 16):  # 16B write granularity
                o = swz_off(r, row_bytes, slot)
                assert 0 <= o < rows * row_bytes, (r, slot, o)
                assert o % 16 == 0
                seen.add(o)
        assert len(seen) == rows * (row_bytes // 16)


@given(st.integers(min_value=0, max_value=10**6),
       st.sampled_from(["", "m", "k", "K", "M", "G", "Ki", "Mi", "Gi"]))
@settings(max_examples=200)
def test_parse_quantity_total(n, suf):
    v = parse_quantity(f"{n}{suf}")
    assert v >= 0
    scale = {"": 1, "m": 1e-3, "k": 1e3, "K": 1e3, "M": 1e6, "G": 1e9,
             "Ki": 2**10, "Mi": 2**20, "Gi": 2**30}[suf]
    assert abs(v - n * scale) <= 1e-6 * max(1.0, n * scale)


@given(st.integers(min_value=0, max_value=300))
@settings(max_examples=300)
def test_exit_code_classifier_total_and_consistent(code):
    """Every exit code classifies without raising, and the documented
    classes hold: 128-255 retryable EXCEPT the permanent signal set;
    1-127 permanent except 0 (handled upstream as success)."""
    r = fo.exit_code_retryable(code)
    assert isinstance(r, bool)
    if code in (130, 137, 138, 143):
        assert r
    if code in (1, 2, 126, 127, 128, 139):
        assert not r


@given(st.integers(min_value=0, max_value=8),
       st.integers(min_value=0, max_value=8),
       st.integers(min_value=0, max_value=4))
@settings(max_examples=100)
def test_crd_round_trip_replicas(workers, gpus, cpus):
    from torch_on_k8s_amd.controlplane.api import (SchedulingPolicy,
                                                   TaskSpec, TaskType,
                                                   TorchJob, set_defaults)
    from torch_on_k8s_amd.controlplane.jobspec import (job_from_dict,
                                                       job_to_crd_dict)
    tasks = {TaskType.MASTER: TaskSpec(replicas=1, gpus_per_task=gpus,
                                       cpus_per_task=float(cpus))}
    if workers:
        tasks[TaskType.WORKER] = TaskSpec(replicas=workers,
                                          gpus_per_task=gpus)
    job = set_defaults(TorchJob(name="p", tasks=tasks,
                                scheduling=SchedulingPolicy(queue="q")))
    job2 = job_from_dict(job_to_crd_dict(job))
    assert job2.total_gpus() == job.total_gpus()
    assert job2.total_replicas() == job.total_replicas()
    assert abs(job2.total_resources()["cpu"] -
               job.total_resources()["cpu"]) < 1e-9


@given(st.text(min_size=0, max_size=40))
@settings(max_examples=300)
def test_job_name_validation_is_filesystem_safe(name):
    """Property: any name that survives set_defaults stays inside a
    directory when joined (no separators, no '..', non-empty) — the
    invariant the spool/status/jobs layout depends on."""
    import os.path
    from torch_on_k8s_amd.controlplane.api import (TaskSpec, TaskType,
                                                   TorchJob, set_defaults)
    try:
        job = set_defaults(TorchJob(
            name=name, tasks={TaskType.MASTER: TaskSpec(replicas=1)}))
    except ValueError:
        return  # rejected: nothing to check
    n = job.name
    assert n and "/" not in n and "\\" not in n and n not in (".", "..")
    joined = os.path.normpath(os.path.join("/base", n))
    assert joined.startswith("/base/") or joined == "/base/" + n


@given(st.integers(min_value=-4, max_value=12),
       st.integers(min_value=0, max_value=8))
@settings(max_examples=60)
def test_allocate_never_overgrants(n, preallocated):
    """Property: allocate(n) either returns exactly n free slots or
    raises; the free count never drops by more than n."""
    import pytest
    from torch_on_k8s_amd.controlplane.node import NodeState
    node = NodeState(num_gpus=8)
    if preallocated:
        node.allocate(min(preallocated, 8), "other")
    free_before = len(node.free_slots)
    if n < 0:
        with pytest.raises(ValueError):
            node.allocate(n, "j")
        assert len(node.free_slots) == free_before
    elif n > free_before:
        with pytest.raises(RuntimeError):
            node.allocate(n, "j")
        assert len(node.free_slots) == free_before
    else:
        got = node.allocate(n, "j")
        assert len(got) == n
        assert len(node.free_slots) == free_before - n

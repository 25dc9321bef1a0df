"""Round-2 coordinator/controller hardening tests (r1 VERDICT weak
#3/#4/#5): quota lifecycle (forget on admission), resource-map quotas,
WRR stability under queue churn, and a daemon soak that holds memory
bounds over hundreds of short jobs."""
from __future__ import annotations

import time

from torch_on_k8s_amd.controlplane.api import (SchedulingPolicy, TaskPhase,
                                               TaskSpec, TaskType, TorchJob,
                                               set_defaults)
from torch_on_k8s_amd.controlplane.controller import (ControllerConfig,
                                                      JobController)
from torch_on_k8s_amd.controlplane.coordinator import (
    Coordinator, Queue, QueueUnit, QuotaPlugin, WeightedRoundRobinSelector)
from torch_on_k8s_amd.controlplane.node import NodeState
from torch_on_k8s_amd.controlplane.runtime import FakeRuntime


def mk_job(name, workers=1, master=True, queue="t", **kw):
    tasks = {}
    if master:
        tasks[TaskType.MASTER] = TaskSpec(replicas=1)
    if workers:
        tasks[TaskType.WORKER] = TaskSpec(replicas=workers)
    return TorchJob(name=name, tasks=tasks,
                    scheduling=SchedulingPolicy(queue=queue), **kw)


def mk_stack(num_gpus=8, quotas=None):
    node = NodeState(num_gpus=num_gpus)
    rt = FakeRuntime()
    ctl = JobController(node, rt, ControllerConfig())
    coord = Coordinator(dequeue_fn=ctl.reconcile,
                        tenant_usage_fn=ctl.tenant_resource_usage,
                        quotas=quotas, default_quota=num_gpus)
    ctl.coordinator = coord
    return ctl, coord, node, rt


# ---------------------------------------------------------------------------
# quota lifecycle: forget() on admission (no 60s double-count)
# ---------------------------------------------------------------------------
def test_sequential_same_tenant_admission_inside_ttl():
    """Job A (4 GPUs) admitted, then job B (4 GPUs) of the SAME tenant
    must be admittable immediately: A's assumed quota is forgotten once
    its tasks exist (r1: it double-counted for 60s and stalled B)."""
    ctl, coord, node, rt = mk_stack(num_gpus=8, quotas={"t": 8})
    a = ctl.create_job(mk_job("a", workers=3))   # 4 tasks x 1 GPU
    assert coord.schedule_once() is not None     # dequeues A
    ctl.reconcile(a)  # DAG: workers start once master is Running
    # A fully admitted -> assumed quota must be gone
    assert a.uid not in coord.quota._assumed
    b = ctl.create_job(mk_job("b", workers=3))
    out = coord.schedule_once()
    assert out is not None and out.name == "b"
    ctl.reconcile(b)
    # both jobs' tasks exist
    assert len(ctl.handles["a"]) == 4 and len(ctl.handles["b"]) == 4


def test_quota_forget_on_job_delete():
    ctl, coord, node, rt = mk_stack(num_gpus=2, quotas={"t": 2})
    a = ctl.create_job(mk_job("a", workers=1))
    coord.schedule_once()
    ctl.reconcile(a)  # second pass: workers created after master Running
    assert a.uid not in coord.quota._assumed
    ctl.delete_job("a")
    b = ctl.create_job(mk_job("b", workers=1))
    assert coord.schedule_once() is not None
    ctl.reconcile(b)
    assert len(ctl.handles["b"]) == 2


# ---------------------------------------------------------------------------
# resource-map quotas (reference covers full ResourceQuota lists)
# ---------------------------------------------------------------------------
def test_resource_map_quota_filters_cpu_and_mem():
    quota = QuotaPlugin(quotas={"t": {"gpu": 8, "cpu": 16, "memory_mb": 1000}})
    job = set_defaults(TorchJob(
        name="j", tasks={TaskType.WORKER: TaskSpec(
            replicas=2, gpus_per_task=1, cpus_per_task=4,
            mem_mb_per_task=300)},
        scheduling=SchedulingPolicy(queue="t")))
    qu = QueueUnit(job)
    # fits: 2 gpu, 8 cpu, 600 mb
    assert quota.filter(qu, {"gpu": 0, "cpu": 0, "memory_mb": 0})
    # cpu exhausted
    assert not quota.filter(qu, {"gpu": 0, "cpu": 9, "memory_mb": 0})
    # memory exhausted
    assert not quota.filter(qu, {"gpu": 0, "cpu": 0, "memory_mb": 500})
    # legacy int in_use means gpu only
    assert quota.filter(qu, 6)
    assert not quota.filter(qu, 7)


def test_int_quota_means_gpu_only():
    quota = QuotaPlugin(quotas={"t": 4})
    job = set_defaults(TorchJob(
        name="j", tasks={TaskType.WORKER: TaskSpec(
            replicas=4, gpus_per_task=1, cpus_per_task=100)},
        scheduling=SchedulingPolicy(queue="t")))
    assert quota.filter(QueueUnit(job), 0)      # cpu not quota-tracked
    assert not quota.filter(QueueUnit(job), 1)  # gpu would exceed


def test_assumed_quota_ttl_expiry_still_works():
    quota = QuotaPlugin(quotas={"t": 4})
    quota.ASSUME_TTL = 0.05
    job = set_defaults(mk_job("a", workers=3))
    quota.pre_dequeue(QueueUnit(job))
    b = QueueUnit(set_defaults(mk_job("b", workers=3)))
    assert not quota.filter(b, 0)
    time.sleep(0.08)
    assert quota.filter(b, 0)


# ---------------------------------------------------------------------------
# WRR under queue churn (keyed by tenant, not index)
# ---------------------------------------------------------------------------
def _mk_queue(tenant, jobs, workers):
    q = Queue(tenant)
    for i in range(jobs):
        q.add(QueueUnit(set_defaults(mk_job(
            f"{tenant}{i}", workers=workers, queue=tenant))))
    return q


def test_wrr_cursor_survives_queue_removal():
    """Removing a queue between calls must not shift the cursor onto a
    different tenant's slot (r1: index-based state mis-pointed)."""
    qa = _mk_queue("a", 1, 3)   # weight 4
    qb = _mk_queue("b", 1, 3)   # weight 4
    qc = _mk_queue("c", 1, 3)   # weight 4
    sel = WeightedRoundRobinSelector()
    first = sel.next([qa, qb, qc]).tenant
    assert first == "a"
    # tenant a disappears; the cursor was at a -> next must be b (the
    # next name in order), not a skipped/doubled pick
    second = sel.next([qb, qc]).tenant
    assert second == "b"
    third = sel.next([qb, qc]).tenant
    assert third == "c"
    # a returns; cursor at c -> wraps to a
    fourth = sel.next([qa, qb, qc]).tenant
    assert fourth == "a"


def test_wrr_fairness_with_churn():
    """Fairness holds (roughly proportional) while queues come and go."""
    sel = WeightedRoundRobinSelector()
    picks = {"a": 0, "b": 0}
    for round_i in range(600):
        qa = _mk_queue("a", 1, 5)   # weight 6
        qb = _mk_queue("b", 1, 2)   # weight 3
        queues = [qa, qb]
        if round_i % 3 == 0:        # transient tenant churns in and out
            queues.append(_mk_queue(f"x{round_i}", 1, 0))
        t = sel.next(queues).tenant
        if t in picks:
            picks[t] += 1
    total = picks["a"] + picks["b"]
    # a:b should be ~2:1
    assert 0.55 < picks["a"] / total < 0.78, picks


# ---------------------------------------------------------------------------
# daemon soak: hundreds of short jobs leave no residue
# ---------------------------------------------------------------------------
def test_soak_no_unbounded_growth():
    ctl, coord, node, rt = mk_stack(num_gpus=8)
    for i in range(300):
        name = f"s{i}"
        job = ctl.create_job(mk_job(name, workers=1, queue=f"ten{i % 7}"))
        coord.schedule_once()
        ctl.reconcile(job)
        # finish both tasks
        for key in list(ctl.handles.get(name, {})):
            rt.set_phase(key, TaskPhase.SUCCEEDED, exit_code=0)
        ctl.reconcile(job)
        assert job.status.phase is not None
        ctl.delete_job(name)
    # all per-job state released
    assert not ctl.jobs and not ctl.handles
    assert not ctl._ports and not ctl._first_task_ts
    assert not ctl._events_by_job
    assert not ctl._event_last
    assert len(ctl.events) <= ctl.events.maxlen
    # coordinator: no tenant-name or assumed-quota leak
    assert not coord.queues
    assert not coord.quota._assumed
    assert not coord._index
    # every GPU slot returned
    assert len(node.free_slots) == 8


def test_adopted_job_bypasses_admission_queue():
    """Manager-restart wedge regression: an adopted job was already
    admitted in its previous life — re-queueing it would deadlock (the
    quota filter counts its OWN live GPUs against the tenant)."""
    node = NodeState(num_gpus=8)
    rt = FakeRuntime()
    ctl = JobController(node, rt, ControllerConfig())
    coord = Coordinator(dequeue_fn=ctl.reconcile,
                        tenant_usage_fn=ctl.tenant_resource_usage,
                        quotas={"t": 8}, default_quota=8)
    ctl.coordinator = coord
    # a previous manager left an 8-GPU gang running (full tenant quota)
    rt.adoptable["big"] = [
        {"pid": 1, "task_type": "master", "index": 0, "generation": 1,
         "gpu_slots": [0]},
    ] + [{"pid": 2 + i, "task_type": "worker", "index": i, "generation": 1,
          "gpu_slots": [1 + i]} for i in range(7)]
    job = ctl.create_job(mk_job("big", workers=7))
    # adopted straight into Running territory, never queued
    assert not coord.is_queuing(job.uid)
    assert len(ctl.handles["big"]) == 8
    assert len(node.free_slots) == 0
    assert any(e.reason == "TaskAdopted" for e in ctl.events)


def test_adoption_churn_conserves_slots():
    """Randomized manager-crash/adopt cycles: across repeated 'crashes'
    (new controller + node over the surviving gang), GPU slots are
    conserved, handles are never duplicated, and every job still runs
    to completion."""
    import random
    rng = random.Random(42)
    for trial in range(10):
        num_gpus = 8
        rt = FakeRuntime()
        ctl = JobController(NodeState(num_gpus=num_gpus), rt,
                            ControllerConfig())
        jobs = []
        for j in range(rng.randint(1, 3)):
            job = ctl.create_job(mk_job(f"c{trial}-{j}",
                                        workers=rng.randint(0, 2),
                                        queue=f"q{j}"))
            jobs.append(job)
        for _ in range(3):
            for job in jobs:
                ctl.reconcile(job)

        for _crash in range(rng.randint(1, 3)):
            # snapshot live handles as adoption records
            recs = {}
            for name, hs in ctl.handles.items():
                recs[name] = [
                    {"pid": 10_000 + i, "task_type": h.task_type.value,
                     "index": h.index, "generation": h.generation,
                     "gpu_slots": list(h.gpu_slots)}
                    for i, h in enumerate(hs.values()) if not h.finished]
            # "crash": new controller + node + runtime
            rt = FakeRuntime()
            rt.adoptable = recs
            ctl = JobController(NodeState(num_gpus=num_gpus), rt,
                                ControllerConfig())
            jobs = [ctl.create_job(mk_job(j.name,
                                          workers=j.tasks.get(
                                              TaskType.WORKER,
                                              TaskSpec(replicas=0)).replicas,
                                          queue=j.scheduling.queue))
                    for j in jobs]
            # no duplicate handles per (job, type, index)
            for name, hs in ctl.handles.items():
                assert len(hs) == len({h.key for h in hs.values()})
            for _ in range(3):
                for job in jobs:
                    ctl.reconcile(job)
            used = sum(len(h.gpu_slots) for hs in ctl.handles.values()
                       for h in hs.values())
            assert used + len(ctl.node.free_slots) == num_gpus

        # drive everything to completion
        for _ in range(6):
            for key, h in list(rt.tasks.items()):
                if not h.finished:
                    rt.set_phase(key, TaskPhase.SUCCEEDED, exit_code=0)
            for job in jobs:
                ctl.reconcile(job)
        for job in jobs:
            assert job.status.phase is not None, job.name
        assert len(ctl.node.free_slots) == num_gpus, \
            (trial, ctl.node.alloc)


def test_priority_class_name_resolution():
    """SchedulingPolicy.priorityClassName resolves against the manager's
    priority-class table when `priority` is unset; an explicit priority
    wins over the class (reference plugins/priority.go:48-85)."""
    from torch_on_k8s_amd.controlplane.coordinator import PriorityPlugin

    plug = PriorityPlugin({"high-priority": 100, "low": -5})
    j_class = set_defaults(mk_job("by-class"))
    j_class.scheduling.priority_class_name = "high-priority"
    j_both = set_defaults(mk_job("both"))
    j_both.scheduling.priority = 7
    j_both.scheduling.priority_class_name = "high-priority"
    j_unknown = set_defaults(mk_job("unknown-class"))
    j_unknown.scheduling.priority_class_name = "nonexistent"
    assert plug.score(QueueUnit(j_class)) == 100.0
    assert plug.score(QueueUnit(j_both)) == 7.0     # explicit wins
    assert plug.score(QueueUnit(j_unknown)) == 0.0  # unknown class -> 0


def test_priority_class_orders_admission():
    """Two jobs in one tenant queue: the one whose priorityClassName maps
    to the larger value is admitted first."""
    admitted = []
    co = Coordinator(dequeue_fn=lambda job: admitted.append(job.name),
                     tenant_usage_fn=lambda t: 0, default_quota=100,
                     priority_classes={"gold": 50, "bronze": 1})
    lo = set_defaults(mk_job("lo", queue="t1"))
    lo.scheduling.priority_class_name = "bronze"
    hi = set_defaults(mk_job("hi", queue="t1"))
    hi.scheduling.priority_class_name = "gold"
    co.enqueue_or_update(lo)
    co.enqueue_or_update(hi)
    co.schedule_once()
    assert admitted[0] == "hi"


def test_spot_preemption_uses_priority_class():
    """A gang blocked on GPUs preempts spot replicas of a job whose
    priorityClassName resolves LOWER, even with no explicit priority set
    on either job (pod.go:592-603 overlay + priority.go resolution)."""
    from torch_on_k8s_amd.controlplane.api import SpotTaskSpec
    from torch_on_k8s_amd.controlplane.controller import ControllerConfig

    node = NodeState(num_gpus=2)
    rt = FakeRuntime()
    ctl = JobController(node, rt, ControllerConfig(
        priority_classes={"gold": 50, "bronze": 1}))
    lo = set_defaults(mk_job("lo-spot", workers=2, master=False))
    lo.tasks[TaskType.WORKER].spot = SpotTaskSpec(num_spot_replicas=2)
    lo.scheduling.priority_class_name = "bronze"
    ctl.create_job(lo)
    ctl.reconcile(lo)
    assert len(node.free_slots) == 0
    hi = set_defaults(mk_job("hi-gang", workers=2, master=False))
    hi.scheduling.priority_class_name = "gold"
    ctl.create_job(hi)
    for _ in range(4):
        ctl.reconcile(hi)
    assert rt.killed, "gold-class gang should preempt bronze spot tasks"


def test_quota_reject_emits_events():
    """Quota rejection must be visible (the reference's flow-controlled
    quota-plugin events, quota.go:59): a request larger than the whole
    tenant quota -> ExceedsTenantQuota; a merely-busy quota ->
    QuotaPending. Neither job dequeues."""
    events = []
    co = Coordinator(dequeue_fn=lambda job: events.append(("deq", job.name)),
                     tenant_usage_fn=lambda t: 6,
                     quotas={"t1": 8},
                     event_fn=lambda j, r, m: events.append((r, j.name)))
    giant = set_defaults(mk_job("giant", workers=15, queue="t1"))  # 16 GPUs
    co.enqueue_or_update(giant)
    co.schedule_once()
    assert ("ExceedsTenantQuota", "giant") in events
    assert ("deq", "giant") not in events
    events.clear()
    busy = set_defaults(mk_job("busy", workers=3, queue="t1"))  # 4 > 8-6
    co.enqueue_or_update(busy)
    co.schedule_once()
    assert ("QuotaPending", "busy") in events
    assert ("deq", "busy") not in events


def test_quota_reject_event_reaches_manager_status(tmp_path):
    """Through the Manager wiring: the Warning event lands in the job's
    event ring (and hence its published status)."""
    from torch_on_k8s_amd.manager import Manager
    import os as _os
    import yaml as _yaml
    mgr = Manager(str(tmp_path), num_gpus=2, sync_period=0.05,
                  quotas={"default": 2})
    doc = {"metadata": {"name": "too-big"},
           "spec": {"tasks": {"master": {"replicas": 1},
                              "worker": {"replicas": 7}}}}  # 8 > 2
    with open(_os.path.join(mgr.spool, "too-big.yaml"), "w") as f:
        _yaml.safe_dump(doc, f)
    import time as _t
    deadline = _t.time() + 10
    while _t.time() < deadline:
        mgr.step()
        if any(e.reason == "ExceedsTenantQuota"
               for e in mgr.controller.events_for("too-big")):
            break
        _t.sleep(0.02)
    assert any(e.reason == "ExceedsTenantQuota"
               for e in mgr.controller.events_for("too-big"))


def test_gang_unsatisfiable_event():
    """A gang larger than the whole node warns GangUnsatisfiable instead
    of the generic waiting event (waiting can never help)."""
    node = NodeState(num_gpus=2)
    ctl = JobController(node, FakeRuntime())
    big = set_defaults(mk_job("big", workers=7))  # 8 GPUs on a 2-GPU node
    ctl.create_job(big)
    ctl.reconcile(big)
    assert any(e.reason == "GangUnsatisfiable"
               for e in ctl.events_for("big"))
    # a merely-busy gang still gets the Normal waiting event
    ok1 = set_defaults(mk_job("ok1", workers=1))  # 2 GPUs
    ctl.create_job(ok1)
    ctl.reconcile(ok1)  # admits, node now full
    ok2 = set_defaults(mk_job("ok2", workers=1))
    ctl.create_job(ok2)
    ctl.reconcile(ok2)
    assert any(e.reason == "GangNotAdmitted"
               for e in ctl.events_for("ok2"))
    assert not any(e.reason == "GangUnsatisfiable"
                   for e in ctl.events_for("ok2"))


def test_sigterm_ignoring_task_keeps_slots_until_reaped():
    """A deleted job's task that survives the graceful SIGTERM (it is
    checkpointing, or stuck) still occupies its physical GPUs: the
    slots must stay HELD until the process really exits, with SIGKILL
    escalation after kill_grace_seconds — releasing at delete time
    would hand a busy GPU to the next gang."""
    from torch_on_k8s_amd.controlplane.api import TaskPhase
    from torch_on_k8s_amd.controlplane.controller import ControllerConfig

    class StubbornRuntime(FakeRuntime):
        def kill(self, h, grace=True):
            self.killed.append((h.key, grace))
            if not grace:  # only SIGKILL fells it, and only on poll
                h._dying = True

        def poll(self, h):
            if getattr(h, "_dying", False):
                h.phase = TaskPhase.FAILED
                h.exit_code = 137
                return h
            return super().poll(h)

    node = NodeState(num_gpus=2)
    rt = StubbornRuntime()
    ctl = JobController(node, rt, ControllerConfig(kill_grace_seconds=0.2))
    job = set_defaults(mk_job("stuck", workers=1, master=False))
    ctl.create_job(job)
    ctl.reconcile(job)
    assert len(node.free_slots) == 1
    ctl.delete_job(job.name)
    # SIGTERM ignored: the slot is NOT back yet
    assert len(node.free_slots) == 1
    assert ctl._reaping
    ctl.reap_pass()  # inside the grace window: still held
    assert len(node.free_slots) == 1
    time.sleep(0.25)
    ctl.reap_pass()  # past deadline: escalates to SIGKILL
    assert ((job.name, TaskType.WORKER, 0), False) in rt.killed
    ctl.reap_pass()  # process now gone: slot released
    assert len(node.free_slots) == 2
    assert not ctl._reaping
    # a new gang can use the GPU only now
    nxt = set_defaults(mk_job("next", workers=2, master=False))
    ctl.create_job(nxt)
    ctl.reconcile(nxt)
    assert len(node.free_slots) == 0


def test_policy_none_straggler_reaped_on_natural_exit():
    """CleanPodPolicy None: a straggler left running at job completion
    (the GPU-less AIMaster sidecar — Succeeded requires master+workers
    done) is tracked by the reap list WITHOUT a kill deadline, never
    killed, and drained when it exits on its own."""
    from torch_on_k8s_amd.controlplane.api import (CleanPodPolicy,
                                                   JobConditionType,
                                                   RunPolicy, TaskPhase,
                                                   TaskSpec)

    node = NodeState(num_gpus=2)
    rt = FakeRuntime()
    ctl = JobController(node, rt)
    job = mk_job("straggle", workers=0)
    job.tasks[TaskType.AIMASTER] = TaskSpec(replicas=1)
    job.run_policy = RunPolicy(clean_task_policy=CleanPodPolicy.NONE)
    set_defaults(job)
    ctl.create_job(job)
    for _ in range(3):
        ctl.reconcile(job)
    rt.set_phase(("straggle", TaskType.MASTER, 0), TaskPhase.SUCCEEDED, 0)
    for _ in range(3):
        ctl.reconcile(job)
    assert job.status.phase == JobConditionType.SUCCEEDED
    assert not rt.killed                      # None policy: no kill
    assert any(h.task_type == TaskType.AIMASTER and d is None
               for h, d in ctl._reaping)      # polled, no SIGKILL clock
    ctl.reap_pass()
    assert ctl._reaping                       # still alive: still tracked
    rt.set_phase(("straggle", TaskType.AIMASTER, 0), TaskPhase.SUCCEEDED, 0)
    ctl.reap_pass()
    assert not ctl._reaping                   # drained on natural exit


def test_per_task_termination_grace_overrides_default():
    """terminationGracePeriodSeconds (pod-spec analog) overrides the
    manager's kill_grace_seconds for that task's SIGKILL escalation."""
    from torch_on_k8s_amd.controlplane.api import TaskPhase
    from torch_on_k8s_amd.controlplane.controller import ControllerConfig
    from torch_on_k8s_amd.controlplane.jobspec import job_from_dict

    class StubbornRuntime(FakeRuntime):
        def kill(self, h, grace=True):
            self.killed.append((h.key, grace))
            if not grace:
                h.phase = TaskPhase.FAILED
                h.exit_code = 137

    # CRD dialect carries the grace in the pod template
    job = job_from_dict({"metadata": {"name": "graceful"}, "spec": {
        "torchTaskSpecs": {"Master": {"numTasks": 1, "template": {"spec": {
            "terminationGracePeriodSeconds": 0.05,
            "containers": [{"name": "torch", "resources": {
                "limits": {"amd.com/gpu": 1}}}]}}}}}})
    assert job.tasks[TaskType.MASTER].termination_grace_seconds == 0.05

    node = NodeState(num_gpus=1)
    rt = StubbornRuntime()
    ctl = JobController(node, rt, ControllerConfig(kill_grace_seconds=9999))
    ctl.create_job(job)
    ctl.reconcile(job)
    ctl.delete_job(job.name)
    assert len(node.free_slots) == 0  # held while SIGTERM pending
    time.sleep(0.1)
    ctl.reap_pass()  # per-task 0.05s grace elapsed despite 9999 default
    assert any(not g for _, g in rt.killed), "never escalated to SIGKILL"
    ctl.reap_pass()
    assert len(node.free_slots) == 1

"""Prometheus metrics parity (reference pkg/metrics/metrics.go +
pkg/coordinator/core/metrics.go):

  counters  torch_on_k8s_amd_jobs_{created,deleted,successful,failed,restarted}
  gauges    torch_on_k8s_amd_jobs_{running,pending}
            torch_on_k8s_amd_tenant_queue_jobs_pending_count{queue}
  histograms job first-task / all-tasks launch delay

Served over HTTP with prometheus_client (metrics/server.go:28-37 analog).
"""
from __future__ import annotations

import time

try:
    from prometheus_client import (Counter, Gauge, Histogram,
                                   start_http_server)
    HAVE_PROM = True
except ImportError:  # pragma: no cover
    HAVE_PROM = False


class JobMetrics:
    _created = None  # class-level so repeated instantiation in tests works

    def __init__(self, registry=None):
        self._job_created_ts: dict[str, float] = {}
        if not HAVE_PROM:
            self._counts = {}
            return
        if JobMetrics._created is None:
            kw = {"registry": registry} if registry is not None else {}
            JobMetrics._created = {
                "created": Counter("torch_on_k8s_amd_jobs_created_total",
                                   "jobs created", **kw),
                "deleted": Counter("torch_on_k8s_amd_jobs_deleted_total",
                                   "jobs deleted", **kw),
                "successful": Counter("torch_on_k8s_amd_jobs_successful_total",
                                      "jobs succeeded", **kw),
                "failed": Counter("torch_on_k8s_amd_jobs_failed_total",
                                  "jobs failed", **kw),
                "restarted": Counter("torch_on_k8s_amd_jobs_restarted_total",
                                     "task restarts", **kw),
                "queue_pending": Gauge(
                    "torch_on_k8s_amd_tenant_queue_jobs_pending_count",
                    "pending jobs per queue", ["queue"], **kw),
                "first_delay": Histogram(
                    "torch_on_k8s_amd_job_first_task_launch_delay_seconds",
                    "job create -> first task start", **kw),
                "all_delay": Histogram(
                    "torch_on_k8s_amd_job_all_tasks_launch_delay_seconds",
                    "job create -> job running", **kw),
            }
        self.m = JobMetrics._created

    # -- recorder API ---------------------------------------------------
    def job_created_at(self, job_name: str):
        self._job_created_ts[job_name] = time.time()

    def _inc(self, name):
        if HAVE_PROM:
            self.m[name].inc()
        else:
            self._counts[name] = self._counts.get(name, 0) + 1

    def created(self):
        self._inc("created")

    def deleted(self):
        self._inc("deleted")

    def succeeded(self):
        self._inc("successful")

    def failed(self):
        self._inc("failed")

    def restarted(self):
        self._inc("restarted")

    def first_task_delay(self, job_name: str):
        t0 = self._job_created_ts.get(job_name)
        if t0 is not None and HAVE_PROM:
            self.m["first_delay"].observe(time.time() - t0)

    def all_tasks_delay(self, job_name: str):
        t0 = self._job_created_ts.get(job_name)
        if t0 is not None and HAVE_PROM:
            self.m["all_delay"].observe(time.time() - t0)

    def set_queue_depth(self, queue: str, depth: int):
        if HAVE_PROM:
            self.m["queue_pending"].labels(queue=queue).set(depth)


def start_metrics_server(port: int = 8443):
    """HTTP /metrics endpoint (reference metrics/server.go:28-37)."""
    if HAVE_PROM:
        start_http_server(port)

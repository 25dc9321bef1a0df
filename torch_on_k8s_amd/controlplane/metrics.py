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
                # data-plane gauges fed from the trainers' structured
                # metrics files (the reference has no data-plane view at
                # all — it scrapes worker stdout with a regex)
                "job_tokens_per_s": Gauge(
                    "torch_on_k8s_amd_job_tokens_per_second",
                    "training throughput per job", ["job"], **kw),
                "job_step": Gauge(
                    "torch_on_k8s_amd_job_step",
                    "latest optimizer step per job", ["job"], **kw),
                "job_loss": Gauge(
                    "torch_on_k8s_amd_job_loss",
                    "latest training loss per job", ["job"], **kw),
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

    def set_training_metrics(self, job: str, rec: dict):
        """Publish a trainer metrics.json record (step/loss/tokens_per_s)."""
        if not HAVE_PROM:
            return
        if rec.get("tokens_per_s") is not None:
            self.m["job_tokens_per_s"].labels(job=job).set(
                rec["tokens_per_s"])
        if rec.get("step") is not None:
            self.m["job_step"].labels(job=job).set(rec["step"])
        if rec.get("loss") is not None:
            self.m["job_loss"].labels(job=job).set(rec["loss"])

    def remove_job(self, job: str):
        if not HAVE_PROM:
            return
        for g in ("job_tokens_per_s", "job_step", "job_loss"):
            try:
                self.m[g].remove(job)
            except KeyError:
                pass


def start_metrics_server(port: int = 8443):
    """HTTP /metrics endpoint (reference metrics/server.go:28-37)."""
    if HAVE_PROM:
        start_http_server(port)

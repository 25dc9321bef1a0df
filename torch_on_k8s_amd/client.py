"""Client SDK — the typed-clientset analog (reference client/ is a
generated Go clientset/informer/lister tree; here the API surface is the
manager's spool/status directories, so the client is a thin typed
wrapper over them: apply/delete/get/list/wait)."""
from __future__ import annotations

import json
import os
import time


class TorchJobClient:
    def __init__(self, workdir: str):
        self.workdir = workdir
        self.spool = os.path.join(workdir, "spool")
        self.status_dir = os.path.join(workdir, "status")
        os.makedirs(self.spool, exist_ok=True)

    def apply(self, spec: "str | dict", name: str | None = None) -> str:
        """Submit a TorchJob YAML/dict (kubectl apply analog)."""
        if isinstance(spec, dict):
            import yaml
            name = name or spec.get("metadata", {}).get("name", "torchjob")
            text = yaml.safe_dump(spec)
        else:
            text = spec
            if name is None:
                import yaml
                name = yaml.safe_load(text).get("metadata", {}).get(
                    "name", "torchjob")
        # name becomes a spool filename: refuse anything k8s-invalid
        # (controlplane.api.set_defaults applies the same rule at parse)
        import re
        if not re.fullmatch(r"[a-zA-Z0-9_]([-a-zA-Z0-9._]{0,251}"
                            r"[a-zA-Z0-9_])?", name or ""):
            raise ValueError(f"invalid job name {name!r}")
        tmp = os.path.join(self.spool, f".{name}.tmp")
        with open(tmp, "w") as f:
            f.write(text)
        os.replace(tmp, os.path.join(self.spool, f"{name}.yaml"))
        return name

    def delete(self, name: str) -> bool:
        try:
            os.unlink(os.path.join(self.spool, f"{name}.yaml"))
            return True
        except FileNotFoundError:
            pass
        # a hand-dropped spool file may use any filename: find the one
        # whose metadata.name owns this job
        import yaml
        try:
            files = os.listdir(self.spool)
        except OSError:
            return False
        for f in files:
            if not f.endswith((".yaml", ".yml", ".json")):
                continue
            path = os.path.join(self.spool, f)
            try:
                with open(path) as fh:
                    doc = yaml.safe_load(fh) or {}
            except (OSError, yaml.YAMLError):
                continue
            if (doc.get("metadata") or {}).get("name") == name:
                try:
                    os.unlink(path)
                    return True
                except OSError:
                    return False
        return False

    def get(self, name: str) -> dict | None:
        try:
            with open(os.path.join(self.status_dir, f"{name}.json")) as f:
                return json.load(f)
        except (OSError, ValueError):
            return None

    def list(self) -> list:
        try:
            return sorted(f[:-5] for f in os.listdir(self.status_dir)
                          if f.endswith(".json"))
        except OSError:
            return []

    def wait(self, name: str, phases=("Succeeded", "Failed"),
             timeout: float = 600, poll: float = 0.5) -> dict:
        t0 = time.time()
        while time.time() - t0 < timeout:
            st = self.get(name)
            if st and st.get("phase") in phases:
                return st
            time.sleep(poll)
        raise TimeoutError(f"job {name} did not reach {phases}; "
                           f"last status: {self.get(name)}")

    def logs(self, name: str, task: str = "master", index: int = 0,
             tail: int | None = None) -> str:
        """Task process log (kubectl logs analog; the runtime writes
        one file per task under the job state dir)."""
        path = os.path.join(self.workdir, "jobs", name,
                            f"{name}-{task}-{index}.log")
        try:
            with open(path) as f:
                text = f.read()
        except OSError:
            return ""
        if tail is not None:
            text = "\n".join(text.splitlines()[-tail:])
        return text

    def describe(self, name: str) -> str:
        """Human-readable status rendering (kubectl describe analog):
        phase, tasks, conditions with ages, recent events, elastic
        state — everything the status schema carries."""
        st = self.get(name)
        if st is None:
            return f"torchjob {name!r} not found"
        now = time.time()

        def age(ts):
            if not ts:
                return "-"
            d = max(0.0, now - ts)
            if d < 120:
                return f"{d:.0f}s"
            if d < 7200:
                return f"{d / 60:.0f}m"
            return f"{d / 3600:.1f}h"

        lines = [f"Name:        {st.get('name', name)}",
                 f"Phase:       {st.get('phase')}",
                 f"Generation:  {st.get('generation')}",
                 f"Restarts:    {st.get('restartCount')}"]
        if st.get("modelVersion"):
            lines.append(f"ModelVersion: {st['modelVersion']}")
        e = st.get("elastic")
        if e:
            lines.append(f"Elastic:     replicas={e.get('currentReplicas')} "
                         f"last={e.get('lastReplicas')} "
                         f"condition={e.get('elasticCondition')} "
                         f"continue={e.get('continue')}")
        lines.append("Tasks:")
        for t, c in (st.get("tasks") or {}).items():
            lines.append(f"  {t:10s} active={c.get('active', 0)} "
                         f"succeeded={c.get('succeeded', 0)} "
                         f"failed={c.get('failed', 0)}")
        lines.append("Conditions:")
        for c in st.get("conditions") or []:
            lines.append(f"  {c.get('type', ''):12s} "
                         f"{age(c.get('ts')):>6s}  {c.get('reason', '')}")
        evs = st.get("events") or []
        lines.append("Events:")
        for ev in evs[-10:]:
            lines.append(f"  {ev.get('type', ''):8s} "
                         f"{age(ev.get('ts')):>6s}  "
                         f"{ev.get('reason', ''):24s} "
                         f"{ev.get('message', '')}")
        if not evs:
            lines.append("  <none>")
        return "\n".join(lines)

    # -- model registry view (Model/ModelVersion read API analog) ------
    def _registry(self):
        from torch_on_k8s_amd.controlplane.modelregistry import (
            ModelRegistry, StorageProvider)
        reg = ModelRegistry(StorageProvider(
            os.path.join(self.workdir, "models")))
        reg.reindex()
        return reg

    def list_models(self) -> dict:
        """{model: {latest, versions: [..]}} from the OCI artifact store."""
        reg = self._registry()
        return {m.name: {"latest": m.latest_version,
                         "versions": sorted(m.versions)}
                for m in reg.models.values()}

    def extract_model(self, ref: str, dest: str) -> str:
        """Unpack model:version (OCI layout) into dest; returns the
        model rootfs path (podman-run analog)."""
        model, _, version = ref.partition(":")
        reg = self._registry()
        if not version:
            m = reg.models.get(model)
            version = m.latest_version if m else None
        if not version:
            raise KeyError(f"no built version for {model}")
        return reg.extract(model, version, dest)


def main():
    """kubectl-style CLI over the manager workdir:
        python -m torch_on_k8s_amd.client apply job.yaml
        python -m torch_on_k8s_amd.client get|delete|wait <name>
        python -m torch_on_k8s_amd.client list
    """
    import argparse
    ap = argparse.ArgumentParser(prog="torch-on-k8s-amd-client")
    ap.add_argument("--workdir", default="/tmp/torch-on-k8s-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)
    p_apply = sub.add_parser("apply")
    p_apply.add_argument("file")
    p_val = sub.add_parser(
        "validate", help="parse a manifest (either dialect) without "
                         "submitting; prints the canonical CRD form")
    p_val.add_argument("file")
    for c in ("get", "delete", "wait", "describe"):
        pc = sub.add_parser(c)
        pc.add_argument("name")
    sub.add_parser("list")
    p_logs = sub.add_parser("logs")
    p_logs.add_argument("name")
    p_logs.add_argument("--task", default="master")
    p_logs.add_argument("--index", type=int, default=0)
    p_logs.add_argument("--tail", type=int, default=None)
    sub.add_parser("models")
    p_ex = sub.add_parser("extract")
    p_ex.add_argument("ref", help="model[:version] (default: latest)")
    p_ex.add_argument("dest")
    args = ap.parse_args()

    cli = TorchJobClient(args.workdir)
    if args.cmd == "apply":
        with open(args.file) as f:
            name = cli.apply(f.read())
        print(f"torchjob/{name} applied")
    elif args.cmd == "validate":
        from torch_on_k8s_amd.controlplane.jobspec import (job_from_yaml,
                                                           job_to_crd_dict)
        import sys as _sys
        import yaml as _yaml
        with open(args.file) as f:
            text = f.read()
        try:
            job = job_from_yaml(text)
        except (ValueError, KeyError, TypeError) as e:
            print(f"INVALID: {e}", file=_sys.stderr)
            raise SystemExit(1)
        print(f"torchjob/{job.name} valid "
              f"({job.total_replicas()} tasks, {job.total_gpus()} GPUs)")
        print(_yaml.safe_dump(job_to_crd_dict(job), sort_keys=False), end="")
    elif args.cmd == "get":
        st = cli.get(args.name)
        print(json.dumps(st, indent=2) if st else f"not found: {args.name}")
    elif args.cmd == "describe":
        print(cli.describe(args.name))
    elif args.cmd == "delete":
        ok = cli.delete(args.name)
        print(f"torchjob/{args.name} {'deleted' if ok else 'not found'}")
    elif args.cmd == "wait":
        st = cli.wait(args.name)
        print(json.dumps(st, indent=2))
    elif args.cmd == "list":
        for n in cli.list():
            st = cli.get(n) or {}
            print(f"{n:30s} {st.get('phase')}")
    elif args.cmd == "logs":
        print(cli.logs(args.name, task=args.task, index=args.index,
                       tail=args.tail))
    elif args.cmd == "models":
        print(json.dumps(cli.list_models(), indent=2))
    elif args.cmd == "extract":
        print(cli.extract_model(args.ref, args.dest))


if __name__ == "__main__":
    main()

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
    for c in ("get", "delete", "wait"):
        pc = sub.add_parser(c)
        pc.add_argument("name")
    sub.add_parser("list")
    args = ap.parse_args()

    cli = TorchJobClient(args.workdir)
    if args.cmd == "apply":
        with open(args.file) as f:
            name = cli.apply(f.read())
        print(f"torchjob/{name} applied")
    elif args.cmd == "get":
        st = cli.get(args.name)
        print(json.dumps(st, indent=2) if st else f"not found: {args.name}")
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


if __name__ == "__main__":
    main()

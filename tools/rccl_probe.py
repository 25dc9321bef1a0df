#!/usr/bin/env python3
"""Probe: can RCCL run 2 ranks on ONE GPU (r1 VERDICT next-#2 asks for
an N=2-ranks-on-1-GPU RCCL smoke "if the runtime permits, else document
why not")? NCCL/RCCL normally rejects two ranks on one device in a
communicator; this probe records the actual behavior on this stack.

Run on a 1-GPU box:  python tools/rccl_probe.py
Prints PROBE_OK / PROBE_FAIL <reason> and exits 0 either way.
"""
from __future__ import annotations

import os
import sys


def rank_main(rank: int):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": "2", "LOCAL_RANK": "0",
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29755",
    })
    import datetime
    import torch
    import torch.distributed as dist
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=2,
                            timeout=datetime.timedelta(seconds=60))
    t = torch.ones(1024, device="cuda") * (rank + 1)
    dist.all_reduce(t)
    ok = torch.allclose(t, torch.full_like(t, 3.0))
    dist.destroy_process_group()
    sys.exit(0 if ok else 1)


def main():
    if len(sys.argv) > 1:
        rank_main(int(sys.argv[1]))
        return
    import subprocess
    procs = [subprocess.Popen([sys.executable, __file__, str(r)],
                              stdout=subprocess.PIPE,
                              stderr=subprocess.STDOUT, text=True)
             for r in range(2)]
    outs = []
    try:
        for p in procs:
            out, _ = p.communicate(timeout=120)
            outs.append((p.returncode, out))
    except subprocess.TimeoutExpired:
        for p in procs:
            p.kill()
        print("PROBE_FAIL timeout (ranks hung in init/collective)")
        return
    if all(rc == 0 for rc, _ in outs):
        print("PROBE_OK 2 RCCL ranks on one GPU all-reduced correctly")
    else:
        tail = " | ".join(o.strip().splitlines()[-1] if o.strip() else ""
                          for _, o in outs)
        print(f"PROBE_FAIL rcs={[rc for rc, _ in outs]}: {tail[:500]}")


if __name__ == "__main__":
    main()

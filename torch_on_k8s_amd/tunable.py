"""hipBLASLt/rocBLAS GEMM tuning via PyTorch TunableOp.

The projection GEMMs go through hipBLASLt (library GEMMs only; fused ops
are handwritten HIP). TunableOp picks the best solution per GEMM shape;
results are tuned offline once (tools/tune_gemms.sh), committed under
tunableop/, and loaded read-only at runtime.

Must be called BEFORE `import torch`.
"""
import glob
import os

TUNED_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tunableop")


def setup_tunableop(tuning: bool = False) -> bool:
    """Enable TunableOp. tuning=False loads committed results only (no
    runtime tuning cost); returns True if enabled."""
    if os.environ.get("TOK_DISABLE_TUNABLEOP") == "1":
        return False
    results = glob.glob(os.path.join(TUNED_DIR, "tunableop_results*.csv"))
    if not (tuning or results):
        return False
    os.makedirs(TUNED_DIR, exist_ok=True)
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1" if tuning else "0")
    os.environ.setdefault(
        "PYTORCH_TUNABLEOP_FILENAME",
        os.path.join(TUNED_DIR, "tunableop_results%d.csv"))
    return True

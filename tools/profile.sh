#!/bin/bash
# rocprofv3 recipes for this repo (run on an MI355X box via gpurun).
# Counter collection must NOT be combined with sys/runtime/hip trace
# domains (suspected node-crasher on this pool) — keep the two modes
# below separate.
set -e
export TMPDIR=/tmp
cd /tmp
OUT=${OUT:-/root/repo/gpurun_out/prof_$(date +%s 2>/dev/null || echo run)}
MODE=${1:-stats}
shift || true
CMD=${@:-"python /root/repo/bench.py --steps 2 --warmup 1"}

case "$MODE" in
  stats)   # per-kernel timing breakdown -> <OUT>/**/*_results.db (rocpd)
    rocprofv3 --kernel-trace --stats -d "$OUT" -- $CMD ;;
  pmc)     # counters only (never add trace domains to this mode)
    rocprofv3 --kernel-trace --pmc SQ_LDS_BANK_CONFLICT SQ_INSTS_MFMA \
      SQ_INSTS_VALU SQ_WAVE_CYCLES SQ_BUSY_CYCLES -d "$OUT" -- $CMD ;;
  *) echo "usage: profile.sh [stats|pmc] [command...]"; exit 2 ;;
esac
echo "wrote $OUT  (query top_kernels / counters_collection in the .db;"
echo " commit summaries to /root/repo/profiles/)"

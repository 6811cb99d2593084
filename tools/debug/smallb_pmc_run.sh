#!/bin/bash
# PMC capture of the smallb scan: v2 (KAKVEDA_SMALLB=2) vs v4 (default).
# Parses rocprofv3 --list-avail robustly, picks L2/cache counters, runs
# both variants under --pmc --kernel-trace only (no trace domains).
set -u
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
rocprofv3 --list-avail > gpurun_out/pmc_avail_raw.txt 2>&1
CTRS=$(python3 - <<'EOF'
import re
txt = open("gpurun_out/pmc_avail_raw.txt").read()
names = set(re.findall(r"\b(TCC_[A-Z0-9_]+|TCP_[A-Z0-9_]+|SQ_WAVES)\b", txt))
want = []
for w in ("TCC_HIT_sum", "TCC_MISS_sum", "TCC_REQ_sum", "TCC_HIT", "TCC_MISS",
          "TCC_REQ", "TCP_TCC_READ_REQ_sum", "TCP_TOTAL_CACHE_ACCESSES_sum",
          "SQ_WAVES"):
    if w in names and len(want) < 4:
        want.append(w)
print(" ".join(want))
EOF
)
echo "counters: [$CTRS]"
if [ -z "$CTRS" ]; then
  echo "no cache counters found; aborting"
  exit 0
fi
for v in 2 4; do
  if [ "$v" = 2 ]; then export KAKVEDA_SMALLB=2; else unset KAKVEDA_SMALLB; fi
  timeout 280 rocprofv3 --pmc $CTRS --kernel-trace \
    -d "gpurun_out/pmc_sb_$v" -- python tools/debug/smallb_pmc_workload.py \
    > "gpurun_out/pmc_sb_$v.log" 2>&1
  grep -m1 csum "gpurun_out/pmc_sb_$v.log" || tail -3 "gpurun_out/pmc_sb_$v.log"
done

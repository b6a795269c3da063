#!/bin/bash
# Round-2 call 28: validate the new all-path scan determinism tests
# (run twice) + fresh post-race-fix rocprof stats for the fleet step.
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
exec > >(tee gpurun_out/r2_call28.log) 2>&1

timeout 240 python -m pytest tests/test_ops_gpu.py -q -m gpu -k deterministic 2>&1 | tail -2
timeout 240 python -m pytest tests/test_ops_gpu.py -q -m gpu -k deterministic 2>&1 | tail -1

export TMPDIR=/tmp
( cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats \
  -d "$GRAFT_REPO_ROOT/gpurun_out/prof_r2final" -- \
  python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 1 --warmup 1 \
  --machines-per-gpu 60 > "$GRAFT_REPO_ROOT/gpurun_out/prof_r2final.log" 2>&1 )
tail -1 gpurun_out/prof_r2final.log

# extract top kernels from the rocpd sqlite
DB=$(ls gpurun_out/prof_r2final/*.db 2>/dev/null | head -1)
if [ -n "$DB" ]; then
  python - "$DB" <<'PY'
import sqlite3, sys
con = sqlite3.connect(sys.argv[1])
try:
    rows = con.execute("""
      SELECT s.display_name, COUNT(*) n,
             SUM(d.end-d.start)/1e6 total_ms,
             AVG(d.end-d.start)/1e3 avg_us
      FROM rocpd_kernel_dispatch d
      JOIN rocpd_info_kernel_symbol s ON d.kernel_id = s.id
      GROUP BY s.display_name ORDER BY total_ms DESC LIMIT 20""").fetchall()
except Exception as e:
    print("query failed:", e); rows = []
tot = sum(r[2] for r in rows)
for name, n, ms, us in rows:
    print(f"{ms:9.2f} ms {100*ms/max(tot,1e-9):5.1f}% {n:6d}x {us:8.1f} us  {name[:90]}")
PY
fi

#!/bin/bash
# rocprofv3 kernel-trace+stats over a short bench run (run on the GPU box).
set -x
REPO=/root/repo
mkdir -p "$REPO/gpurun_out/prof"
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof" -o bench_prof -- \
  bash -c "cd $REPO && python bench.py --steps 10 --warmup 3 > gpurun_out/bench_prof_run.json 2> gpurun_out/bench_prof_run.err"
echo "prof rc=$?"
grep '^# ' "$REPO/gpurun_out/bench_prof_run.err" || true
ls "$REPO/gpurun_out/prof/" | head

#!/bin/bash
# rocprofv3 kernel-trace+stats over a short bench run (run on the GPU box).
# Produces a compact kernel-time summary in gpurun_out/prof_summary.txt and
# removes the bulky trace files so the gpurun merge stays under 64 MiB.
set -x
REPO=/root/repo
STEPS=${STEPS:-10}
rm -rf "$REPO/gpurun_out/prof"
mkdir -p "$REPO/gpurun_out/prof"
cd /tmp && export TMPDIR=/tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv \
  -d "$REPO/gpurun_out/prof" -o bench_prof -- \
  bash -c "cd $REPO && python bench.py --steps $STEPS --warmup 3 > gpurun_out/bench_prof_run.json 2> gpurun_out/bench_prof_run.err"
echo "prof rc=$?"
grep '^# ' "$REPO/gpurun_out/bench_prof_run.err" || true
ls -la "$REPO/gpurun_out/prof/"
python - <<'EOF'
import csv, glob, os
repo = '/root/repo'
out = []
for f in glob.glob(repo + '/gpurun_out/prof/*stats*.csv'):
    out.append('== %s ==' % os.path.basename(f))
    with open(f) as fh:
        rows = list(csv.DictReader(fh))
    key = 'TotalDurationNs' if rows and 'TotalDurationNs' in rows[0] else None
    if key:
        rows.sort(key=lambda r: -float(r[key]))
    for r in rows[:40]:
        name = (r.get('Name') or r.get('KernelName') or '?')[:100]
        out.append('%10.3f ms  %6s calls  avg %8.1f us  %s' % (
            float(r.get('TotalDurationNs', 0)) / 1e6,
            r.get('Calls', '?'),
            float(r.get('AverageNs', 0)) / 1e3,
            name))
with open(repo + '/gpurun_out/prof_summary.txt', 'w') as fh:
    fh.write('\n'.join(out))
print('\n'.join(out[:45]))
EOF
rm -rf "$REPO/gpurun_out/prof"

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 540 python -m pytest tests -x -q -m gpu > gpurun_out/gputests25.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests25.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke25.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/smoke25.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b25.json 2> gpurun_out/b25.err
echo "bench rc=$?"
grep -h '"value"' gpurun_out/b25.json
timeout 300 python tools/bench_geister.py --actors 8192 --workers 8 > gpurun_out/g25.log 2>&1
echo "geister rc=$?"
grep -h '"value"' gpurun_out/g25.log

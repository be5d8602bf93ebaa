set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 600 python -m pytest tests -x -q -m gpu > gpurun_out/gputests10.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests10.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke10.log 2>&1
echo "smoke rc=$?"; tail -2 gpurun_out/smoke10.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b10_driver_1.json 2> gpurun_out/b10_driver_1.err
echo "driver1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b10_driver_2.json 2> gpurun_out/b10_driver_2.err
echo "driver2 rc=$?"
timeout 420 python bench.py > gpurun_out/b10_default.json 2> gpurun_out/b10_default.err
echo "default rc=$?"
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g10.log 2>&1
echo "geister rc=$?"
timeout 600 python tools/learning_check.py 1000 > gpurun_out/learn10_1k.log 2>&1
echo "learn1k rc=$?"; tail -3 gpurun_out/learn10_1k.log
grep -h '"value"' gpurun_out/b10_*.json gpurun_out/g10.log

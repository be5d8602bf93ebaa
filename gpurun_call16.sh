set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 python tools/bench_geister.py --actors 4096 --workers 8 > gpurun_out/g16_4096.log 2>&1
echo "g4096 rc=$?"
timeout 300 python tools/bench_geister.py --actors 2048 --workers 12 > gpurun_out/g16_2048w12.log 2>&1
echo "g2048w12 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 8192 > gpurun_out/b16_e8192.json 2> gpurun_out/b16_e8192.err
echo "e8192 rc=$?"
timeout 600 python -m pytest tests -x -q -m gpu > gpurun_out/gputests16.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests16.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke16.log 2>&1
echo "smoke rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b16_driver.json 2> gpurun_out/b16_driver.err
echo "driver rc=$?"
grep -h '"value"' gpurun_out/g16_*.log gpurun_out/b16_*.json

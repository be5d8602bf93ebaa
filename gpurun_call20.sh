set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 python tools/bench_geister.py --actors 2048 --workers 8 > gpurun_out/g20_traj_2048.log 2>&1
echo "traj2048 rc=$?"
timeout 300 python tools/bench_geister.py --actors 8192 --workers 8 > gpurun_out/g20_traj_8192.log 2>&1
echo "traj8192 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b20_1.json 2> gpurun_out/b20_1.err
echo "b1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b20_2.json 2> gpurun_out/b20_2.err
echo "b2 rc=$?"
timeout 420 python tools/learning_check.py 400 > gpurun_out/learn20.log 2>&1
echo "learn rc=$?"; tail -2 gpurun_out/learn20.log
grep -h '"value"' gpurun_out/g20_*.log gpurun_out/b20_*.json

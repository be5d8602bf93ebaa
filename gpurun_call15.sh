set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 30 > gpurun_out/b15_steady_1.json 2> gpurun_out/b15_steady_1.err
echo "steady1 rc=$?"
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 30 > gpurun_out/b15_steady_2.json 2> gpurun_out/b15_steady_2.err
echo "steady2 rc=$?"
timeout 300 python tools/bench_geister.py --actors 2048 --workers 8 > gpurun_out/g15_2048.log 2>&1
echo "g2048 rc=$?"
timeout 300 python tools/bench_geister.py --actors 1024 --workers 12 > gpurun_out/g15_1024w12.log 2>&1
echo "g1024w12 rc=$?"
grep -h '"value"' gpurun_out/b15_*.json gpurun_out/g15_*.log
grep -h "mean_len\|episodes=" gpurun_out/b15_steady_1.err

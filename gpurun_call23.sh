set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 30 > gpurun_out/b23_steady_1.json 2> gpurun_out/b23_steady_1.err
echo "s1 rc=$?"
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 30 > gpurun_out/b23_steady_2.json 2> gpurun_out/b23_steady_2.err
echo "s2 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b23_driver.json 2> gpurun_out/b23_driver.err
echo "d rc=$?"
timeout 300 python tools/bench_geister.py --actors 8192 --workers 8 > gpurun_out/g23_8192.log 2>&1
echo "g rc=$?"
grep -h '"value"' gpurun_out/b23_*.json gpurun_out/g23_8192.log
grep -h "mean_len" gpurun_out/b23_*.err

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g14_async.log 2>&1
echo "g_async rc=$?"
HANDYRL_GEISTER_ASYNC=0 timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g14_sync.log 2>&1
echo "g_sync rc=$?"
timeout 300 python tools/bench_geister.py --actors 1024 --workers 8 > gpurun_out/g14_async_1024.log 2>&1
echo "g_1024 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b14_1.json 2> gpurun_out/b14_1.err
echo "bench1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b14_2.json 2> gpurun_out/b14_2.err
echo "bench2 rc=$?"
timeout 480 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests14.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests14.log
grep -h '"value"' gpurun_out/g14_*.log gpurun_out/b14_*.json
grep -h "illegal sampled" gpurun_out/g14_*.log | head -3
grep -h "actor ms" gpurun_out/b14_1.err

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/e2e_gpu_train_eval.py 3 > gpurun_out/e2e_v3.log 2>&1
echo "e2e rc=$?"; grep -E "TRAIN_DONE|EVAL_DONE|win rate|---agent|total " gpurun_out/e2e_v3.log | tail -12
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g13_async.log 2>&1
echo "g_async rc=$?"
HANDYRL_GEISTER_ASYNC=0 timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g13_sync.log 2>&1
echo "g_sync rc=$?"
timeout 300 python tools/bench_geister.py --actors 1024 --workers 8 > gpurun_out/g13_async_1024.log 2>&1
echo "g_async_1024 rc=$?"
grep -h '"value"' gpurun_out/g13_*.log
timeout 420 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests13.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests13.log

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/e2e_gpu_train_eval.py 3 Geister > gpurun_out/e2e_geister.log 2>&1
echo "e2e_geister rc=$?"; grep -E "TRAIN_DONE|EVAL_DONE|---agent|total " gpurun_out/e2e_geister.log | tail -8
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b21.json 2> gpurun_out/b21.err
echo "b rc=$?"
grep -h '"value"' gpurun_out/b21.json

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/e2e_gpu_train_eval.py 3 > gpurun_out/e2e_v2.log 2>&1
echo "e2e rc=$?"; grep -E "epoch|win rate|TRAIN_DONE|EVAL_DONE|total|---agent" gpurun_out/e2e_v2.log | tail -16
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b12_driver_1.json 2> gpurun_out/b12_driver_1.err
echo "driver1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b12_driver_2.json 2> gpurun_out/b12_driver_2.err
echo "driver2 rc=$?"
grep -h '"value"' gpurun_out/b12_*.json

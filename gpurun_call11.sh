set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/e2e_gpu_train_eval.py 3 > gpurun_out/e2e_train_eval.log 2>&1
echo "e2e rc=$?"; grep -E "epoch|win rate|TRAIN_DONE|EVAL_DONE|total" gpurun_out/e2e_train_eval.log | tail -20
HANDYRL_ENT_REG=0.3 timeout 600 python tools/learning_check.py 1000 > gpurun_out/learn11_ent03_1k.log 2>&1
echo "ent03_1k rc=$?"; tail -3 gpurun_out/learn11_ent03_1k.log

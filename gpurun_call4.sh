set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests4.log 2>&1
echo "gputests rc=$?"; tail -3 gpurun_out/gputests4.log
timeout 240 python tools/rccl_probe.py > gpurun_out/rccl_probe4.log 2>&1
echo "rccl_probe rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b4_svc_w8.json 2> gpurun_out/b4_svc_w8.err
echo "svc_w8 rc=$?"
HANDYRL_ACTOR_PROCS=4 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b4_svc_w4.json 2> gpurun_out/b4_svc_w4.err
echo "svc_w4 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b4_svc_w8_e4096.json 2> gpurun_out/b4_svc_w8_e4096.err
echo "svc_e4096 rc=$?"
HANDYRL_ACTOR_SLOTS=2 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b4_svc_s2.json 2> gpurun_out/b4_svc_s2.err
echo "svc_s2 rc=$?"
HANDYRL_SHM_REGISTER=0 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b4_svc_noreg.json 2> gpurun_out/b4_svc_noreg.err
echo "svc_noreg rc=$?"
timeout 420 python tools/learning_check.py 400 > gpurun_out/learn4.log 2>&1
echo "learning rc=$?"; tail -4 gpurun_out/learn4.log
grep -h '"value"' gpurun_out/b4_*.json

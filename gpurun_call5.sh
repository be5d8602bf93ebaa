set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/svc_graph_probe.py > gpurun_out/svcprobe.log 2>&1
echo "svcprobe rc=$?"; tail -6 gpurun_out/svcprobe.log
timeout 420 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests5.log 2>&1
echo "gputests rc=$?"; tail -3 gpurun_out/gputests5.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b5_core_w8.json 2> gpurun_out/b5_core_w8.err
echo "core_w8 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b5_core_e4096.json 2> gpurun_out/b5_core_e4096.err
echo "core_e4096 rc=$?"
HANDYRL_ACTOR_PROCS=4 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b5_core_w4.json 2> gpurun_out/b5_core_w4.err
echo "core_w4 rc=$?"
timeout 420 python tools/learning_check.py 400 > gpurun_out/learn5.log 2>&1
echo "learning rc=$?"; tail -4 gpurun_out/learn5.log
grep -h '"value"' gpurun_out/b5_*.json
grep -h "actor ms" gpurun_out/b5_core_w8.err

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python tools/svc_graph_probe.py > gpurun_out/svcprobe6.log 2>&1
echo "svcprobe rc=$?"; grep SVC_PROBE gpurun_out/svcprobe6.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b6_w8_1.json 2> gpurun_out/b6_w8_1.err
echo "w8_1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b6_w8_2.json 2> gpurun_out/b6_w8_2.err
echo "w8_2 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b6_e4096.json 2> gpurun_out/b6_e4096.err
echo "e4096 rc=$?"
timeout 420 python tools/learning_check.py 400 > gpurun_out/learn6.log 2>&1
echo "learning rc=$?"; tail -3 gpurun_out/learn6.log
timeout 300 python tools/tight_parity_probe.py > gpurun_out/tightprobe.log 2>&1
echo "tightprobe rc=$?"; cat gpurun_out/tightprobe.log
timeout 420 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests6.log 2>&1
echo "gputests rc=$?"; tail -3 gpurun_out/gputests6.log
grep -h '"value"' gpurun_out/b6_*.json

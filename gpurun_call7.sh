set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 480 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests7.log 2>&1
echo "gputests rc=$?"; tail -3 gpurun_out/gputests7.log
AMD_SERIALIZE_KERNEL=3 timeout 420 python bench.py --gpus 1 --steps 3 --warmup 1 > gpurun_out/b7_serial.json 2> gpurun_out/b7_serial.err
echo "serial rc=$?"; tail -25 gpurun_out/b7_serial.err
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b7_w8.json 2> gpurun_out/b7_w8.err
echo "w8 rc=$?"
grep -h '"value"' gpurun_out/b7_*.json

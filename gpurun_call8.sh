set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 480 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests8.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests8.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b8_w8_1.json 2> gpurun_out/b8_w8_1.err
echo "w8_1 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b8_w8_2.json 2> gpurun_out/b8_w8_2.err
echo "w8_2 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b8_e4096.json 2> gpurun_out/b8_e4096.err
echo "e4096 rc=$?"
HANDYRL_ACTOR_PROCS=12 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b8_e4096_w12.json 2> gpurun_out/b8_e4096_w12.err
echo "e4096_w12 rc=$?"
timeout 420 python tools/learning_check.py 400 > gpurun_out/learn8.log 2>&1
echo "learning rc=$?"; tail -3 gpurun_out/learn8.log
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g8_fused.log 2>&1
echo "geister rc=$?"
cd /tmp && export TMPDIR=/tmp
timeout 360 rocprofv3 --kernel-trace --stats -d /tmp/prof8 -o prof8 -- bash -c 'cd $GRAFT_REPO_ROOT && HANDYRL_ACTOR_PROCS=0 python bench.py --gpus 1 --steps 10 --warmup 3' > /root/repo/gpurun_out/rocprof8.log 2>&1
echo "rocprof rc=$?"
find /tmp/prof8 -name '*stats*' -o -name '*.csv' | head -5
find /tmp/prof8 \( -name '*stats*.csv' -o -name '*kernel*.csv' \) -exec cp {} /root/repo/gpurun_out/ \; 2>/dev/null
ls /tmp/prof8* 2>/dev/null | head
grep -h '"value"' /root/repo/gpurun_out/b8_*.json
grep -h '"value"' /root/repo/gpurun_out/g8_fused.log

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 python tools/bench_geister.py --actors 2048 --workers 8 > gpurun_out/g19_traj_2048.log 2>&1
echo "traj2048 rc=$?"
HANDYRL_GEISTER_TRAJ=0 timeout 300 python tools/bench_geister.py --actors 2048 --workers 8 > gpurun_out/g19_pipe_2048.log 2>&1
echo "pipe2048 rc=$?"
timeout 300 python tools/bench_geister.py --actors 8192 --workers 8 > gpurun_out/g19_traj_8192.log 2>&1
echo "traj8192 rc=$?"
timeout 480 python -m pytest tests/test_gpu.py -x -q > gpurun_out/gputests19.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/gputests19.log
grep -h '"value"' gpurun_out/g19_*.log

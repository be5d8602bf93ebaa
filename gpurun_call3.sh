set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/gputests3.log 2>&1
echo "gputests rc=$?"
tail -5 gpurun_out/gputests3.log
timeout 240 python tools/rccl_probe.py > gpurun_out/rccl_probe3.log 2>&1
echo "rccl_probe rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b3_traj_1.json 2> gpurun_out/b3_traj_1.err
echo "traj1 rc=$?"
HANDYRL_TRAJ=0 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b3_notraj.json 2> gpurun_out/b3_notraj.err
echo "notraj rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b3_traj_2.json 2> gpurun_out/b3_traj_2.err
echo "traj2 rc=$?"
HANDYRL_ACTOR_SLOTS=2 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b3_traj_s2.json 2> gpurun_out/b3_traj_s2.err
echo "traj_s2 rc=$?"
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g3_fused.log 2>&1
echo "geister_fused rc=$?"
HANDYRL_DRC_FUSED=0 timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/g3_eager.log 2>&1
echo "geister_eager rc=$?"
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof3 -o prof3 -- bash -c 'cd $GRAFT_REPO_ROOT && python bench.py --gpus 1 --steps 10 --warmup 5' > /root/repo/gpurun_out/rocprof3.log 2>&1
echo "rocprof rc=$?"
cp /tmp/prof3/*stats* /root/repo/gpurun_out/ 2>/dev/null || find /tmp/prof3 -name '*.csv' -exec cp {} /root/repo/gpurun_out/ \;
grep -h '"value"' /root/repo/gpurun_out/b3_*.json
grep -h '"value"' /root/repo/gpurun_out/g3_*.log | tail -2

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
for i in 1 2; do
  timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b9_e4096_w8_$i.json 2> gpurun_out/b9_e4096_w8_$i.err
  echo "e4096_w8_$i rc=$?"
  HANDYRL_ACTOR_PROCS=12 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 4096 > gpurun_out/b9_e4096_w12_$i.json 2> gpurun_out/b9_e4096_w12_$i.err
  echo "e4096_w12_$i rc=$?"
  HANDYRL_ACTOR_PROCS=12 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --envs 8192 > gpurun_out/b9_e8192_w12_$i.json 2> gpurun_out/b9_e8192_w12_$i.err
  echo "e8192_w12_$i rc=$?"
done
timeout 300 python tools/learning_check.py 300 > gpurun_out/learn9_fused.log 2>&1
echo "learn_fused rc=$?"; tail -2 gpurun_out/learn9_fused.log
HANDYRL_FUSED_LOSS=0 timeout 300 python tools/learning_check.py 300 > gpurun_out/learn9_eager.log 2>&1
echo "learn_eager rc=$?"; tail -2 gpurun_out/learn9_eager.log
HANDYRL_ENT_REG=0.3 timeout 300 python tools/learning_check.py 300 > gpurun_out/learn9_ent03.log 2>&1
echo "learn_ent03 rc=$?"
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof9 -o prof9 -- bash -c 'cd $GRAFT_REPO_ROOT && python bench.py --gpus 1 --steps 10 --warmup 3' > /root/repo/gpurun_out/rocprof9.log 2>&1
echo "rocprof rc=$?"
find /tmp/prof9 -name '*.csv' | head
find /tmp/prof9 -name '*stats*.csv' -exec cp {} /root/repo/gpurun_out/ \;
grep -h '"value"' /root/repo/gpurun_out/b9_*.json
grep -h "guard_fires\|step  30" /root/repo/gpurun_out/learn9_*.log | tail -8

set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
timeout 360 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof24 -o prof24 -- bash -c 'cd $GRAFT_REPO_ROOT && python bench.py --gpus 1 --steps 10 --warmup 3' > /root/repo/gpurun_out/rocprof24.log 2>&1
echo "rocprof rc=$?"
find /tmp/prof24 -name '*stats*.csv' -exec cp {} /root/repo/gpurun_out/ \;
cd $GRAFT_REPO_ROOT
timeout 600 python tools/learning_check.py 2000 > gpurun_out/learn24_2k.log 2>&1
echo "learn2k rc=$?"; tail -3 gpurun_out/learn24_2k.log
timeout 700 python tools/e2e_gpu_train_eval.py 2 > gpurun_out/e2e_geese24.log 2>&1
echo "e2e rc=$?"; grep -E "TRAIN_DONE|EVAL_DONE" gpurun_out/e2e_geese24.log; grep -A3 "agent 0" gpurun_out/e2e_geese24.log | head -4
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b24.json 2> gpurun_out/b24.err
echo "b rc=$?"
grep -h '"value"' gpurun_out/b24.json

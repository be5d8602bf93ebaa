set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/gputests.log 2>&1
echo "gputests rc=$?"
timeout 240 python tools/rccl_probe.py > gpurun_out/rccl_probe.log 2>&1
echo "rccl_probe rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/bench_default_w5.json 2> gpurun_out/bench_default_w5.err
echo "bench_default_w5 rc=$?"
HANDYRL_SHM_REGISTER=1 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/bench_shmreg_w5.json 2> gpurun_out/bench_shmreg_w5.err
echo "bench_shmreg_w5 rc=$?"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 30 > gpurun_out/bench_default_w30.json 2> gpurun_out/bench_default_w30.err
echo "bench_default_w30 rc=$?"
HANDYRL_SHM_REGISTER=1 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 30 > gpurun_out/bench_shmreg_w30.json 2> gpurun_out/bench_shmreg_w30.err
echo "bench_shmreg_w30 rc=$?"
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 > gpurun_out/geister_batcher.log 2>&1
echo "geister_batcher rc=$?"
timeout 300 python tools/bench_geister.py --actors 512 --workers 8 --device-replay > gpurun_out/geister_devreplay.log 2>&1
echo "geister_devreplay rc=$?"
tail -2 gpurun_out/bench_default_w5.json gpurun_out/bench_shmreg_w5.json gpurun_out/bench_default_w30.json gpurun_out/bench_shmreg_w30.json gpurun_out/rccl_probe.log

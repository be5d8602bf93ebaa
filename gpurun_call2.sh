set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 240 python tools/rccl_probe.py > gpurun_out/rccl_probe2.log 2>&1
echo "rccl_probe rc=$?"
# interleaved A/B: polling (new default) vs sync-ish baseline behavior is
# not directly re-creatable; instead measure polling at several worker
# counts and slots, interleaved, w5 driver regime
for i in 1 2; do
  timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b2_poll_w6_$i.json 2> gpurun_out/b2_poll_w6_$i.err
  echo "poll_w6_$i rc=$?"
  HANDYRL_ACTOR_PROCS=8 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b2_poll_w8_$i.json 2> gpurun_out/b2_poll_w8_$i.err
  echo "poll_w8_$i rc=$?"
  HANDYRL_ACTOR_PROCS=12 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b2_poll_w12_$i.json 2> gpurun_out/b2_poll_w12_$i.err
  echo "poll_w12_$i rc=$?"
  HANDYRL_ACTOR_SLOTS=2 timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/b2_poll_s2_$i.json 2> gpurun_out/b2_poll_s2_$i.err
  echo "poll_s2_$i rc=$?"
done
grep -h '"value"' gpurun_out/b2_*.json | head -20
tail -3 gpurun_out/rccl_probe2.log

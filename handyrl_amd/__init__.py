"""handyrl_amd — an MI355X-native distributed reinforcement-learning framework.

A from-scratch rebuild of the capabilities of DeNA/HandyRL (reference:
/root/reference, MIT license) designed for AMD Instinct MI355X (gfx950):

* learner data-parallelism: one process per GPU, RCCL (torch.distributed
  "nccl" backend on ROCm) gradient all-reduce over xGMI — replaces the
  reference's single-process nn.DataParallel (reference train.py:339-340);
* self-play actors: batched GPU inference over vectorized environments on
  dedicated HIP streams — replaces one-CPU-process-per-environment workers
  (reference worker.py, generation.py);
* hot compute: hand-written CDNA4 HIP kernels (handyrl_amd/ops/src) for the
  off-policy target scans (V-Trace / TD(lambda) / UPGO, reference losses.py),
  masked-softmax action sampling (reference generation.py:53-58) and the
  torus convolution stack of the flagship Hungry Geese network
  (reference envs/kaggle/hungry_geese.py:23-57);
* the outer API (main.py modes, config.yaml schema, .pth checkpoint layout,
  BaseEnvironment contract) is kept compatible with the reference.
"""

__version__ = "0.1.0"

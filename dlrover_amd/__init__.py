"""dlrover_amd — an MI355X-native elastic distributed-training framework.

A from-scratch rebuild of the capabilities of intelligent-machine-learning/dlrover
(reference layout documented in SURVEY.md), designed AMD-first:

- the data plane is PyTorch-ROCm + hand-written HIP/CDNA4 (gfx950) kernels
  (``dlrover_amd.ops``) + RCCL over xGMI (``torch.distributed`` backend "nccl"
  resolves to RCCL on ROCm);
- the control plane (job master, elastic agent, rendezvous, flash checkpoint,
  diagnosis) is pure Python and GPU-agnostic, mirroring the reference's
  behavior (see docstring citations ``ref: <file:line>`` throughout).

Top-level subpackages:
  common/   cross-cutting substrate: RPC messages, node model, shm IPC, storage
  master/   per-job master: rendezvous, node lifecycle, data sharding, diagnosis
  agent/    per-node elastic agent: worker mgmt, ckpt saver, monitors
  trainer/  user-facing: dlrover-run CLI, ElasticTrainer, flash checkpoint
  models/   flagship model families (Llama-3, nanoGPT) built on our HIP ops
  ops/      HIP/CDNA4 kernels (RMSNorm, RoPE, SwiGLU, fused AdamW, attention)
  parallel/ DDP/FSDP helpers over RCCL
"""

__version__ = "0.1.0"

# hang diagnosis: when the elastic agent exports DLROVER_PY_TRACER_DIR,
# every worker that imports this package gains a SIGUSR2 handler dumping
# all-thread Python stacks (diagnosis/py_tracer.py)
import os as _os

if _os.getenv("DLROVER_PY_TRACER_DIR"):
    from dlrover_amd.diagnosis.py_tracer import maybe_install_from_env

    maybe_install_from_env()

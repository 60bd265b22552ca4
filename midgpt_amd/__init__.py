"""midgpt_amd: an MI355X-native GPT pretraining framework.

A from-scratch rebuild of the capabilities of AllanYangZhou/midGPT
(/root/reference) designed for AMD Instinct MI355X (gfx950, CDNA4):

- framework layer: PyTorch-ROCm (replacing JAX/Equinox/optax/orbax)
- hot ops: hand-written HIP kernels on MFMA (flash attention, RoPE,
  RMSNorm, QK-LayerNorm, fused cross-entropy, fused AdamW) in
  ``midgpt_amd/ops/csrc`` — no CUDA compat shims, no Triton
- distributed: explicit RCCL collectives over xGMI (param all-gather +
  grad reduce-scatter), one process per GPU via torch.distributed

The user contract mirrors the reference: ``launch.py --config=<name>
[--rundir] [--debug] [--distributed]``, Python-dataclass configs with the
same field names (reference src/train.py:26-44), the same rundir /
checkpoint layout, and the same metric names.
"""

__version__ = "0.1.0"

from midgpt_amd.config import ExperimentConfig, GPTConfig  # noqa: F401

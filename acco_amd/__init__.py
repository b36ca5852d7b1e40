"""acco_amd — an MI355X-native ACCO training engine.

A from-scratch reimplementation of the capabilities of `edouardoyallon/acco`
("ACCO: Accumulate While You Communicate", arXiv 2406.02613) designed for
AMD Instinct MI355X (gfx950, CDNA4):

- flat bf16 parameter/gradient arenas with model tensors aliasing them
  (replaces the reference's parameters_to_vector view tricks,
  cf. reference trainer_base.py:284-332),
- the ACCO two-round decoupled data-parallel state machine
  (cf. reference trainer_decoupled.py:431-598) on two HIP streams,
- ZeRO-1 sharded fused-AdamW on hand-written gfx950 HIP kernels,
- RCCL reduce-scatter / all-gather over xGMI with multi-bucket launches,
- hand-written CDNA4 kernels for the model hot ops (RMSNorm, RoPE, SwiGLU,
  causal flash attention, fused cross-entropy),
- DDP and DPU baseline modes (cf. reference trainer_decoupled.py:605-833).
"""

__version__ = "0.1.0"

from acco_amd.engine.trainer import DecoupledTrainer  # noqa: F401

"""distributedtraining_amd — MI355X-native decentralized weight-delta training.

A from-scratch framework with the capabilities of bit-current/DistributedTraining
("hivetrain"): miners running local SGD on a shared base model, validators
scoring per-miner weight deltas by held-out loss improvement, and an averager
merging deltas with meta-learned per-parameter weights.

MI355X-first design decisions (vs. the reference's CUDA-free Python + HF-Hub
transport — see SURVEY.md):

* The transformer fwd/bwd hot path runs on hand-written CDNA4 (gfx950) HIP
  kernels: fused causal flash attention, fused LayerNorm/RMSNorm, GELU,
  fused log-softmax cross-entropy, fused AdamW — exposed via
  ``torch.autograd.Function`` (plain projection GEMMs go to hipBLASLt
  through PyTorch, per the MFMA/library split).
* Bulk tensor transport (the reference's HuggingFace-Hub git/LFS repos,
  hivetrain/hf_manager.py) is RCCL collectives over xGMI when running as one
  process per GPU, with a content-hash file store retained as the
  single-process / cross-node fallback (reference: LocalHFManager,
  hf_manager.py:200-241).
* Coordination (the reference's Bittensor chain, btt_connector.py) is an
  in-process/file registry (reference models this itself:
  LocalBittensorNetwork, btt_connector.py:530-671).
* All per-parameter Python dict loops (delta compute, merge, grad_W) are
  fused single-kernel passes over one flat parameter buffer sized for
  288 GB HBM3E.

Initialization is explicit — importing this package has no side effects
(the reference executes config parsing + chain dialing at import time,
training_manager.py:22-24; we deliberately do not).
"""

__version__ = "0.2.0"
__spec_version__ = 100 * 0 + 10 * 2 + 0  # reference: hivetrain/__init__.py:1-10

from . import config  # noqa: F401

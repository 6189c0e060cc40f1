"""r2d2_amd — MI355X-native R2D2 distributed-RL trainer.

Built from scratch for AMD Instinct MI355X (gfx950): PyTorch-ROCm host code,
hand-written HIP/CDNA4 kernels for the training hot path, RCCL over xGMI for
multi-learner data parallelism.  API surface mirrors the ZiyuanMa/R2D2
reference (worker.py / train.py / config.py / model.py) — see SURVEY.md.
"""

__version__ = "0.1.0"

from . import config  # noqa: F401

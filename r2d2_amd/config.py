"""Flat-module hyperparameter config with dataclass-validated presets.

Keeps the ergonomics of the reference's flat ``config.py`` module
(reference: /root/reference/config.py:1-37 — a bare module of constants imported
by name everywhere) while backing it with a validated dataclass and named
presets for the five BASELINE.json configs.  Module-level names stay importable
(``from r2d2_amd import config; config.batch_size``); ``apply(preset)`` or
``apply(**overrides)`` rewrites them in place.
"""

from dataclasses import dataclass, field, asdict
from typing import Optional, Tuple
import sys

# ---------------------------------------------------------------------------
# Dataclass schema
# ---------------------------------------------------------------------------


@dataclass
class Config:
    # environment ---------------------------------------------------------
    game_name: str = "MsPacman"                 # reference: config.py:2
    env_type: str = "synthetic"                 # 'synthetic' | 'cartpole'
    obs_shape: Tuple[int, int, int] = (4, 84, 84)  # (C, H, W); reference uses (1,84,84)
    action_dim: int = 9                          # MsPacman action space
    frame_skip: int = 4

    # training ------------------------------------------------------------
    lr: float = 1e-4                             # reference: config.py:4
    eps: float = 1e-3                            # Adam epsilon; reference: config.py:5
    grad_norm: float = 40.0                      # reference: worker.py:364
    batch_size: int = 64                         # reference: config.py:7
    learning_starts: int = 50000                 # reference: config.py:8
    save_interval: int = 500                     # reference: worker.py:380
    target_net_update_interval: int = 2000       # reference: worker.py:376
    gamma: float = 0.997                         # reference: config.py:11
    prio_exponent: float = 0.9                   # alpha; reference: config.py:12
    importance_sampling_exponent: float = 0.6    # beta; reference: config.py:13
    training_steps: int = 100000                 # reference: config.py:15
    buffer_capacity: int = 2_000_000             # transitions; reference: config.py:16
    max_episode_steps: int = 27000
    actor_update_interval: int = 400             # reference: config.py:18 (hardcoded at worker.py:560)
    block_length: int = 400                      # reference: config.py:19

    # loss: the reference uses MSE (worker.py:290); the paper uses Huber.
    # 'huber' is the default here, 'mse' is the reference-compat flag.
    loss_fn: str = "huber"
    huber_kappa: float = 1.0

    # exploration ---------------------------------------------------------
    num_actors: int = 8                          # reference: config.py:21
    # batched-inference actor driver (VectorActor): one process stepping all
    # num_actors envs in lockstep with one network forward per tick (on
    # actor_device), instead of one process per actor
    vector_actors: bool = False
    actor_device: str = "cpu"                    # VectorActor inference device
    base_eps: float = 0.4                        # reference: config.py:22
    alpha_eps: float = 7.0                       # reference: config.py:23
    eval_eps: float = 0.001

    # sequence layout -----------------------------------------------------
    burn_in_steps: int = 40                      # reference: config.py:27
    learning_steps: int = 40                     # reference: config.py:28
    forward_steps: int = 5                       # n-step n; reference: config.py:29

    # model ---------------------------------------------------------------
    encoder: str = "nature"                      # 'mlp' | 'nature' | 'impala'
    hidden_dim: int = 512                        # reference: config.py:33
    mlp_hidden: int = 128                        # CartPole MLP width

    # runtime -------------------------------------------------------------
    device: str = "cuda"
    dtype: str = "bf16"                          # compute dtype on GPU ('bf16'|'fp32')
    use_hip_kernels: bool = True                 # HIP path on GPU; eager is the CPU/golden path
    gpu_replay: bool = True                      # GPU-resident block store + sum-tree
    log_interval: int = 10                       # seconds; reference: config.py (log_interval)
    metrics_path: Optional[str] = None           # JSONL metrics emit (None = console only)
    # host-replay persistence (elastic resume; the reference never saves
    # replay contents — SURVEY §5).  Set a path to snapshot the ReplayBuffer
    # every replay_snapshot_interval seconds; train(resume=...) restores it.
    replay_snapshot_path: Optional[str] = None
    replay_snapshot_interval: float = 300.0      # seconds between snapshots
    # host-replay batch assembler threads (the slice copies release the
    # GIL; 2 saturates the pipeline when CPU actors share the box)
    assemble_threads: int = 2
    batch_queue_size: int = 8
    amp: bool = True

    # priority mixture eta: prio = eta*max + (1-eta)*mean   (worker.py:268-276)
    prio_eta: float = 0.9
    # value rescale epsilon (worker.py:383-390)
    rescale_eps: float = 1e-3

    def __post_init__(self):
        assert self.encoder in ("mlp", "nature", "impala"), self.encoder
        assert self.loss_fn in ("huber", "mse"), self.loss_fn
        assert self.dtype in ("bf16", "fp32"), self.dtype
        assert self.block_length % self.learning_steps == 0
        assert 0.0 < self.gamma <= 1.0
        assert self.forward_steps >= 1
        assert len(self.obs_shape) in (1, 3)

    # derived -------------------------------------------------------------
    @property
    def seq_len(self) -> int:                    # reference: config.py:30
        return self.burn_in_steps + self.learning_steps + self.forward_steps

    @property
    def seq_per_block(self) -> int:
        return self.block_length // self.learning_steps

    @property
    def num_blocks(self) -> int:                 # reference: worker.py:47
        return self.buffer_capacity // self.block_length


# ---------------------------------------------------------------------------
# Presets — BASELINE.json configs[0..4]
# ---------------------------------------------------------------------------

PRESETS = {
    # configs[0]: plumbing, no GPU
    "cartpole": dict(
        game_name="CartPole", env_type="cartpole", obs_shape=(4,),
        action_dim=2, encoder="mlp", hidden_dim=128, mlp_hidden=128,
        device="cpu", dtype="fp32", use_hip_kernels=False, gpu_replay=False,
        buffer_capacity=40_000, learning_starts=2_000, block_length=40,
        burn_in_steps=8, learning_steps=8, forward_steps=3,
        training_steps=2_000, num_actors=2, amp=False,
    ),
    # configs[1]: 1x MI355X learner bf16, host-pinned replay
    "mspacman": dict(
        game_name="MsPacman", env_type="synthetic", obs_shape=(4, 84, 84),
        action_dim=9, encoder="nature", gpu_replay=False,
    ),
    # configs[2]: GPU-resident prioritized replay, 256 actors
    "mspacman_gpu_replay": dict(
        game_name="MsPacman", env_type="synthetic", obs_shape=(4, 84, 84),
        action_dim=9, encoder="nature", gpu_replay=True, num_actors=256,
        buffer_capacity=4_000_000, vector_actors=True, actor_device="cuda",
    ),
    # configs[3]: 8x data-parallel learners (parallelism degree comes from
    # torchrun's WORLD_SIZE; the preset is otherwise mspacman_gpu_replay)
    "mspacman_dp": dict(
        game_name="MsPacman", env_type="synthetic", obs_shape=(4, 84, 84),
        action_dim=9, encoder="nature", gpu_replay=True, num_actors=256,
        buffer_capacity=4_000_000, vector_actors=True, actor_device="cuda",
    ),
    # configs[4]: IMPALA-deep ResNet encoder
    "seaquest_impala": dict(
        game_name="Seaquest", env_type="synthetic", obs_shape=(4, 84, 84),
        action_dim=18, encoder="impala",
    ),
}


_current = Config(**PRESETS["mspacman"])


def get() -> Config:
    """The live Config object."""
    return _current


def apply(preset: Optional[str] = None, **overrides) -> Config:
    """Apply a named preset and/or field overrides, updating both the Config
    object and this module's flat attribute namespace."""
    global _current
    unknown = [k for k in overrides
               if k not in Config.__dataclass_fields__
               and k not in ("seq_len", "seq_per_block", "num_blocks")]
    if unknown:
        raise TypeError(f"unknown config field(s): {unknown} — a silently "
                        f"dropped override is a misconfigured run")
    base = dict(PRESETS[preset]) if preset else asdict(_current)
    base.update(overrides)
    # drop derived keys if present
    for k in ("seq_len", "seq_per_block", "num_blocks"):
        base.pop(k, None)
    base = {k: v for k, v in base.items() if k in Config.__dataclass_fields__}
    if preset:
        merged = dict(PRESETS[preset]); merged.update(
            {k: v for k, v in overrides.items() if k in Config.__dataclass_fields__})
        _current = Config(**merged)
    else:
        _current = Config(**base)
    _export(_current)
    return _current


def _export(cfg: Config):
    mod = sys.modules[__name__]
    for k, v in asdict(cfg).items():
        setattr(mod, k, v)
    # derived values, flat like the reference's config.py:30
    mod.seq_len = cfg.seq_len
    mod.seq_per_block = cfg.seq_per_block
    mod.num_blocks = cfg.num_blocks


_export(_current)

"""Config dataclass validation + flat-module export semantics
(reference config.py is a bare constants module; ours adds validation,
presets, and override round-trips — SURVEY §5 'Config / flag system')."""

import pytest

from r2d2_amd import config as cfg


def test_presets_cover_baseline_configs():
    for name in ("cartpole", "mspacman", "mspacman_gpu_replay",
                 "mspacman_dp", "seaquest_impala"):
        c = cfg.apply(name)
        assert c.seq_len == (c.burn_in_steps + c.learning_steps
                         + c.forward_steps)
        assert c.block_length % c.learning_steps == 0
    cfg.apply("mspacman")


def test_flat_module_export():
    cfg.apply("mspacman", batch_size=32)
    assert cfg.batch_size == 32
    assert cfg.seq_len == cfg.burn_in_steps + cfg.learning_steps + cfg.forward_steps
    cfg.apply("mspacman")
    assert cfg.batch_size == 64


def test_validation_rejects_bad_values():
    with pytest.raises(AssertionError):
        cfg.apply("mspacman", encoder="transformer")
    with pytest.raises(AssertionError):
        cfg.apply("mspacman", block_length=401)
    with pytest.raises(AssertionError):
        cfg.apply("mspacman", gamma=1.5)
    with pytest.raises(AssertionError):
        cfg.apply("mspacman", loss_fn="l1")
    cfg.apply("mspacman")


def test_override_without_preset_keeps_state():
    cfg.apply("cartpole")
    c = cfg.apply(buffer_capacity=1280, block_length=16, learning_steps=8,
                  burn_in_steps=8)
    assert c.game_name == "CartPole"       # preset retained
    assert c.num_blocks == 1280 // 16
    cfg.apply("mspacman")


def test_apply_rejects_unknown_override():
    """A typo'd override must raise, not silently configure a different
    run (apply() used to drop unknown keys on the floor)."""
    import pytest

    from r2d2_amd import config as cfg

    with pytest.raises(TypeError, match="unknown config field"):
        cfg.apply("cartpole", num_envz=32)
    # derived keys remain accepted (and recomputed)
    c = cfg.apply("cartpole", seq_len=999)
    assert c.seq_len == (c.burn_in_steps + c.learning_steps
                         + c.forward_steps)

import numpy as np
import pytest

from r2d2_amd import config as cfg
from r2d2_amd.worker import LocalBuffer


def make_buffer(action_dim=3, n=2, burn=4, learn=4, block=8, hidden=8, gamma=0.9):
    return LocalBuffer(action_dim, forward_steps=n, burn_in_steps=burn,
                       learning_steps=learn, gamma=gamma, hidden_dim=hidden,
                       block_length=block)


def fill(buf, steps, action_dim=3, hidden=8, reward_fn=lambda t: float(t + 1),
         start=0):
    for t in range(start, start + steps):
        obs = np.full((2, 2), t + 1, dtype=np.uint8)
        q = np.arange(action_dim, dtype=np.float32) + t
        h = np.full((2, hidden), t + 1, dtype=np.float32)
        buf.add(t % action_dim, reward_fn(t), obs, q, h)


def test_block_shapes_and_layout():
    buf = make_buffer()
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    fill(buf, 8)
    block, prios, ep_reward = buf.finish(np.zeros(3, dtype=np.float32))
    assert block.num_sequences == 2
    assert block.obs.shape == (9, 2, 2)           # 0 burn-in + 8 + 1
    assert block.action.shape == (8,)
    assert block.hidden.shape == (2, 2, 8)
    assert list(block.burn_in_steps) == [0, 4]
    assert list(block.learning_steps) == [4, 4]
    assert list(block.forward_steps) == [2, 1]    # bounded by block end
    assert ep_reward is None                      # not done
    assert prios.shape == (2,)
    assert (prios > 0).all()


def test_n_step_reward_and_gamma():
    g, n = 0.5, 2
    buf = make_buffer(n=n, gamma=g)
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    fill(buf, 8, reward_fn=lambda t: 1.0)
    block, _, _ = buf.finish(np.zeros(3, dtype=np.float32))
    # n-step return: 1 + 0.5 = 1.5 except final step (only 1 reward)
    np.testing.assert_allclose(block.n_step_reward[:-1], 1.5)
    np.testing.assert_allclose(block.n_step_reward[-1], 1.0)
    # gamma vector: g^2 until the bootstrap cut, then g^2( wait: decaying )
    np.testing.assert_allclose(block.gamma[:-2], g ** n)
    np.testing.assert_allclose(block.gamma[-2:], [g ** 2, g ** 1])


def test_terminal_gamma_zero_and_episode_reward():
    buf = make_buffer()
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    fill(buf, 6, reward_fn=lambda t: 2.0)
    block, _, ep_reward = buf.finish(None)  # terminal
    assert ep_reward == pytest.approx(12.0)
    assert (block.gamma[-2:] == 0).all()
    assert block.num_sequences == 2
    assert list(block.learning_steps) == [4, 2]


def test_burn_in_carry_over():
    buf = make_buffer()
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    fill(buf, 8)
    b1, _, _ = buf.finish(np.zeros(3, dtype=np.float32))
    assert buf.curr_burn_in_steps == 4            # kept burn_in steps
    fill(buf, 8, start=8)
    b2, _, _ = buf.finish(np.zeros(3, dtype=np.float32))
    assert list(b2.burn_in_steps) == [4, 4]
    assert b2.obs.shape == (13, 2, 2)             # 4 carry + 8 + 1
    # carried prefix is the tail of block 1's obs
    np.testing.assert_array_equal(b2.obs[:5], b1.obs[-5:])


def test_hidden_alignment_burn_in_start():
    """Stored hidden must be the recurrent state at the sequence's burn-in
    START (the fidelity fix over reference worker.py:461)."""
    buf = make_buffer()
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    # hidden added at step t has value t+1 == state at buffer index t+1
    fill(buf, 8)
    b1, _, _ = buf.finish(np.zeros(3, dtype=np.float32))
    # seq 0: burn 0, learning starts at buffer index 0 -> hidden index 0 (zeros)
    np.testing.assert_allclose(b1.hidden[0], 0.0)
    # seq 1: burn 4, learning starts at index 4, burn-in start = 0 -> zeros too
    np.testing.assert_allclose(b1.hidden[1], 0.0)
    fill(buf, 8, start=8)
    b2, _, _ = buf.finish(np.zeros(3, dtype=np.float32))
    # block 2 buffer: carried indices correspond to original steps 4..12
    # seq 0: burn 4 (curr), learning start idx 4, burn-in start idx 0 ->
    #   hidden value == original hidden_buffer[4] == 4 (state after step 3)
    np.testing.assert_allclose(b2.hidden[0], 4.0)
    # seq 1: learning start idx 8, burn 4 -> start idx 4 -> value 8
    np.testing.assert_allclose(b2.hidden[1], 8.0)


def test_priorities_follow_td_magnitude():
    buf = make_buffer()
    buf.reset(np.zeros((2, 2), dtype=np.uint8))
    # rewards all zero, q all zero -> zero priority
    for t in range(8):
        buf.add(0, 0.0, np.zeros((2, 2), dtype=np.uint8),
                np.zeros(3, dtype=np.float32), np.zeros((2, 8), dtype=np.float32))
    _, prios, _ = buf.finish(np.zeros(3, dtype=np.float32))
    np.testing.assert_allclose(prios, 0.0)


def test_full_400_step_block_finish():
    """Regression: a FULL reference-sized block (400 steps, 10 sequences of
    uint8 learning lengths) finishes without overflow and with valid
    priorities (found live via the training demo)."""
    from r2d2_amd import config as cfg
    cfg.apply("mspacman")
    c = cfg.get()
    rng = np.random.default_rng(0)
    lb = LocalBuffer(c.action_dim)
    obs = rng.integers(0, 256, size=tuple(c.obs_shape), dtype=np.uint8)
    lb.reset(obs)
    for t in range(c.block_length):
        lb.add(int(rng.integers(c.action_dim)), float(rng.normal()),
               rng.integers(0, 256, size=tuple(c.obs_shape), dtype=np.uint8),
               rng.normal(size=c.action_dim).astype(np.float32),
               np.zeros((2, c.hidden_dim), dtype=np.float32))
    block, prio, reward = lb.finish(
        rng.normal(size=c.action_dim).astype(np.float32))
    assert block.num_sequences == 10
    assert int(block.learning_steps.sum()) == 400
    assert np.isfinite(prio).all() and prio.shape == (10,)
    cfg.apply("mspacman")


# ---------------------------------------------------------------------------
# property-based invariants (hypothesis): random episode/block interleavings
# ---------------------------------------------------------------------------

from hypothesis import given, settings, strategies as st


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(1, 4),
    burn=st.integers(0, 6),
    learn=st.integers(1, 6),
    spb=st.integers(1, 4),
    cuts=st.lists(st.integers(1, 30), min_size=1, max_size=4),
    done_last=st.booleans(),
    seed=st.integers(0, 10_000),
)
def test_localbuffer_invariants_random(n, burn, learn, spb, cuts, done_last,
                                       seed):
    """For arbitrary (n-step, burn-in, learning, block) geometry and random
    mid-episode block cuts: every finished block satisfies the structural
    invariants the replay/gather stack relies on."""
    rng = np.random.default_rng(seed)
    block_len = learn * spb
    A, H = 3, 4
    buf = LocalBuffer(A, forward_steps=n, burn_in_steps=burn,
                      learning_steps=learn, gamma=0.9, hidden_dim=H,
                      block_length=block_len)
    buf.reset(np.zeros((2,), dtype=np.uint8))
    carried = 0
    for ci, steps in enumerate(cuts):
        steps = min(steps, block_len)
        for t in range(steps):
            buf.add(int(rng.integers(A)), float(rng.normal()),
                    rng.integers(0, 255, size=(2,)).astype(np.uint8),
                    rng.normal(size=A).astype(np.float32),
                    rng.normal(size=(2, H)).astype(np.float32))
        last = ci == len(cuts) - 1
        done = last and done_last
        block, prios, reward = buf.finish(
            None if done else rng.normal(size=A).astype(np.float32))
        S = len(block.action)
        assert S == steps
        # obs rows = carried burn-in + steps + 1
        assert block.obs.shape[0] == block.burn_in_steps[0] + S + 1
        assert block.burn_in_steps[0] == carried
        # sequence layout: learning sums to S, forward ends at the block end
        assert int(block.learning_steps[:block.num_sequences].sum()) == S
        assert block.forward_steps[block.num_sequences - 1] == 1
        assert (block.forward_steps[:block.num_sequences] >= 1).all()
        assert (block.forward_steps[:block.num_sequences] <= n).all()
        # gamma: zero only at terminal steps
        g = block.gamma
        if done:
            assert g[-1] == 0.0
        else:
            assert (g > 0).all()
        assert np.isfinite(prios).all() and (prios >= 0).all()
        assert prios.shape == (spb,)
        # priorities beyond num_sequences are zero (killed tree slots)
        assert (prios[block.num_sequences:] == 0).all()
        # burn-in carry for the next block
        carried = min(burn, carried + S)
        assert reward is None or done

"""ALE wrapper preprocessing: the area resize must match cv2.INTER_AREA
("pixel area relation") semantics for the reference's 210x160 -> 84x84 warp
(/root/reference/environment.py:39-63), computed here against an
independent naive fractional-overlap reference."""

import numpy as np
import pytest

from r2d2_amd.envs.atari import _area_resize_84, _area_weights


def naive_area_resize(img, n_out=84):
    """Independent O(n^4) fractional pixel-area average."""
    h, w = img.shape
    sy, sx = h / n_out, w / n_out
    out = np.empty((n_out, n_out))
    for i in range(n_out):
        for j in range(n_out):
            y0, y1 = i * sy, (i + 1) * sy
            x0, x1 = j * sx, (j + 1) * sx
            acc = 0.0
            for r in range(int(np.floor(y0)), int(np.ceil(y1))):
                wy = min(y1, r + 1) - max(y0, r)
                for c in range(int(np.floor(x0)), int(np.ceil(x1))):
                    wx = min(x1, c + 1) - max(x0, c)
                    acc += wy * wx * img[r, c]
            out[i, j] = acc / (sy * sx)
    return out


def test_weights_are_row_stochastic():
    for n_in in (210, 160, 84, 100):
        w = _area_weights(n_in, 84)
        assert w.shape == (84, n_in)
        np.testing.assert_allclose(w.sum(axis=1), 1.0, atol=1e-12)
        assert (w >= 0).all()


def test_resize_matches_naive_fractional_area_210x160():
    rng = np.random.default_rng(0)
    img = rng.integers(0, 256, size=(210, 160)).astype(np.uint8)
    got = _area_resize_84(img)
    ref = naive_area_resize(img.astype(np.float64))
    ref_u8 = np.clip(np.rint(ref), 0, 255).astype(np.uint8)
    # identical up to the last-bit rounding of the accumulation order
    assert np.abs(got.astype(int) - ref_u8.astype(int)).max() <= 1
    assert (got == ref_u8).mean() > 0.99


def test_resize_preserves_constants_and_mean():
    img = np.full((210, 160), 137, dtype=np.uint8)
    out = _area_resize_84(img)
    assert (out == 137).all()
    rng = np.random.default_rng(1)
    img = rng.integers(0, 256, size=(210, 160)).astype(np.uint8)
    out = _area_resize_84(img)
    # area averaging preserves the global mean exactly (row-stochastic x2)
    assert abs(out.mean() - img.mean()) < 1.0


def test_resize_integer_ratio_case():
    """168x168 -> 84x84 is the exact 2x2 box filter."""
    rng = np.random.default_rng(2)
    img = rng.integers(0, 256, size=(168, 168)).astype(np.uint8)
    out = _area_resize_84(img)
    ref = img.astype(np.float64).reshape(84, 2, 84, 2).mean(axis=(1, 3))
    ref_u8 = np.clip(np.rint(ref), 0, 255).astype(np.uint8)
    np.testing.assert_array_equal(out, ref_u8)


def test_resize_throughput():
    """The per-frame warp must sustain actor-scale rates (>2k frames/s on
    an idle core).  Best-of-5 timing so a transiently loaded machine (the
    CI box may be running other jobs) cannot flake the assert — the
    regression this guards (the original per-pixel Python loop) was 60x
    slower, far outside any contention noise."""
    import time

    from r2d2_amd.envs.atari import limit_blas_threads
    limit_blas_threads()   # the production (actor-process) configuration
    rng = np.random.default_rng(3)
    img = rng.integers(0, 256, size=(210, 160)).astype(np.uint8)
    _area_resize_84(img)  # warm the weight cache
    n = 50
    best = float("inf")
    for _ in range(5):
        t0 = time.perf_counter()
        for _ in range(n):
            _area_resize_84(img)
        best = min(best, time.perf_counter() - t0)
    rate = n / best
    assert rate > 1000, f"{rate:.0f} frames/s (best of 5)"


def test_atari_env_gated():
    """Full AtariEnv contract — runs only where gymnasium+ALE exist."""
    gym = pytest.importorskip("gymnasium")
    try:
        env = __import__("r2d2_amd.envs.atari", fromlist=["AtariEnv"]) \
            .AtariEnv("MsPacman", obs_shape=(1, 84, 84), seed=0)
    except Exception as e:  # ROMs absent
        pytest.skip(f"ALE unavailable: {e}")
    obs = env.reset()
    assert obs.shape == (1, 84, 84) and obs.dtype == np.uint8
    obs, r, done, info = env.step(0)
    assert obs.shape == (1, 84, 84)
    assert isinstance(r, float) and isinstance(done, bool)

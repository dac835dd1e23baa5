"""The examples run and produce sane results (GPU)."""
import pytest

pytestmark = pytest.mark.gpu


def test_monte_carlo_pi():
    from examples.monte_carlo_pi import estimate_pi
    pi = estimate_pi(1 << 22)
    assert abs(pi - 3.14159265) < 0.01


def test_power_iteration():
    from examples.power_iteration import dominant_eig
    lam = dominant_eig(512, iters=25)
    assert abs(lam - 256) < 10   # uniform matrix: lambda ~ n/2


def test_column_stats():
    from examples.column_stats import column_stats
    means, mx, finite, med = column_stats(2048, 256)
    assert finite
    assert abs(float(means.mean()) - 0.5) < 0.01
    assert float(mx.min()) > 0.99
    assert abs(med - 0.5) < 0.05


def test_game_of_life():
    from examples.game_of_life import run
    ok, pop = run(128, 8)
    assert ok, "life diverged from numpy reference"
    assert pop > 0


def test_zscore():
    import numpy as np
    from examples.zscore import zscore
    X, mu, sig, Z = zscore(1024, 128)
    hX = X.collect()
    hZ = Z.collect()
    ref = hX - hX.mean(axis=0, keepdims=True)
    ref = ref / np.sqrt((hX * hX).mean(axis=0, keepdims=True)
                        - hX.mean(axis=0, keepdims=True) ** 2)
    assert np.max(np.abs(hZ - ref)) < 1e-11
    for d in (X, mu, sig, Z):
        d.close()

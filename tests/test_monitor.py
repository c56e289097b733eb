import numpy as np

from persia_amd.core.monitor import DistinctIdMonitor, HyperLogLog


def test_hll_estimates_within_tolerance():
    rng = np.random.default_rng(0)
    for true_n in (100, 10_000, 1_000_000):
        hll = HyperLogLog(p=14)
        signs = rng.integers(0, 2 ** 63, size=true_n, dtype=np.uint64)
        signs = np.unique(signs)
        # feed in chunks with duplicates
        for _ in range(2):
            hll.add_signs(signs)
        est = hll.estimate()
        assert abs(est - len(signs)) / len(signs) < 0.05, (true_n, est)


def test_monitor_per_slot():
    m = DistinctIdMonitor()
    rng = np.random.default_rng(1)
    m.observe("a", rng.integers(0, 1000, size=5000, dtype=np.uint64))
    m.observe("b", np.arange(100, dtype=np.uint64))
    est = m.estimates()
    assert abs(est["a"] - 1000) / 1000 < 0.1
    assert abs(est["b"] - 100) / 100 < 0.1

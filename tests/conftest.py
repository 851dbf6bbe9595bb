import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")


@pytest.fixture
def small_case():
    """20-node BA case with servers/relays/mobiles and link rates set."""
    from multihop_offload_amd import CaseGraph
    rng = np.random.RandomState(42)
    g = CaseGraph(20, t_max=1000, seed=7, gtype="ba")
    g.links_init(50.0, rng=rng)
    g.add_relay(0)
    g.add_relay(1)
    for s in (2, 3, 4):
        g.add_server(s, 300.0)
    for n in range(5, 20):
        g.set_mobile_bw(n, 10.0)
    return g


@pytest.fixture
def jobs_for(small_case):
    from multihop_offload_amd import JobInstance
    rng = np.random.RandomState(1)
    return JobInstance.sample(small_case.mobile_nodes, 0.15, rng)

import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU")


@pytest.fixture
def ray_start_regular():
    """Reference counterpart: python/ray/tests/conftest.py:651."""
    import ray_amd as ray

    ctx = ray.init(num_cpus=4, ignore_reinit_error=True)
    yield ctx
    ray.shutdown()


@pytest.fixture
def ray_start_2_cpus():
    import ray_amd as ray

    ctx = ray.init(num_cpus=2, ignore_reinit_error=True)
    yield ctx
    ray.shutdown()


@pytest.fixture
def ray_start_cluster():
    """Multi-raylet-on-localhost harness (reference:
    python/ray/cluster_utils.py:141)."""
    from ray_amd.cluster_utils import Cluster

    cluster = Cluster()
    yield cluster
    cluster.shutdown()

import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def cpu_cfg():
    """Small, fast CartPole config used by many CPU tests."""
    from distributedes_amd import preset
    return preset("cartpole_es", pop=8, generations=3, episode_steps=40,
                  noise_table_size=1 << 16, device="cpu")

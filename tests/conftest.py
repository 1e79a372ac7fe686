import os
import sys

# Workers must not try to grab real GPUs during CPU tests; individual GPU tests
# override via the gpu marker + their own env.
os.environ.setdefault("MODAL_AMD_NUM_GPUS", "0" if not os.environ.get("MODAL_AMD_FORCE_GPUS") else os.environ["MODAL_AMD_FORCE_GPUS"])
os.environ.setdefault("MODAL_AMD_STATE_DIR", os.path.join(os.path.dirname(__file__), ".test_state"))
os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")
    config.addinivalue_line("markers", "slow: long-running CPU test")


@pytest.fixture(autouse=True, scope="session")
def _cleanup_state():
    yield
    import shutil

    shutil.rmtree(os.environ["MODAL_AMD_STATE_DIR"], ignore_errors=True)


@pytest.fixture
def gpu_env():
    """On a GPU box: clear the CPU-test override so the pool sees real devices."""
    old = os.environ.pop("MODAL_AMD_NUM_GPUS", None)
    yield
    if old is not None:
        os.environ["MODAL_AMD_NUM_GPUS"] = old

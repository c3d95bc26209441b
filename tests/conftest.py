import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an AMD GPU (MI355X) and the HIP extension"
    )


@pytest.fixture(autouse=True)
def _default_dtype_fp64():
    """Reference parity: experiments default to float64."""
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.float64)
    yield
    torch.set_default_dtype(prev)

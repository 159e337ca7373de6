import sys
from pathlib import Path

import pytest
import torch

# repo root importable
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU on this host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def device():
    return "cuda:0" if torch.cuda.is_available() else "cpu"


def make_random_csr(B=64, F_total=10_000, min_f=5, max_f=40, seed=0,
                    device="cpu", binary_vals=True):
    """Random variable-length CSR batch (libffm-shaped)."""
    g = torch.Generator().manual_seed(seed)
    counts = torch.randint(min_f, max_f + 1, (B,), generator=g)
    nnz = int(counts.sum())
    row_ptr = torch.zeros(B + 1, dtype=torch.int32)
    row_ptr[1:] = torch.cumsum(counts, 0).to(torch.int32)
    fids = torch.randint(0, F_total, (nnz,), generator=g, dtype=torch.int32)
    if binary_vals:
        vals = torch.ones(nnz)
    else:
        vals = torch.rand(nnz, generator=g) * 2
    labels = torch.randint(0, 2, (B,), generator=g).float()
    dev = torch.device(device)
    return (row_ptr.to(dev), fids.to(dev), vals.to(dev), labels.to(dev))

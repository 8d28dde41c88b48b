"""CPU-side behavior of the ops layer (no GPU present)."""

import pytest
import torch

from k3samd import ops


def test_missing_native_raises_loudly():
    if ops.native_available():
        pytest.skip("native extension present")
    with pytest.raises(RuntimeError, match="native extension"):
        ops.stream_triad(torch.zeros(4), torch.zeros(4), torch.zeros(4), 1.0)


def test_native_rejects_cpu_tensors():
    if not ops.native_available():
        pytest.skip("native extension not built")
    a = torch.zeros(16)
    with pytest.raises(Exception):
        ops.stream_triad(a, a.clone(), a.clone(), 1.0)

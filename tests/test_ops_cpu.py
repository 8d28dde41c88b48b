"""CPU-side behavior of the ops layer (no GPU present)."""

import pytest
import torch

from k3samd import ops


def test_missing_native_raises_loudly(monkeypatch):
    """No silent fallback: with the extension unimportable, every op raises
    a clear error (simulated by failing the `from k3samd import _C`)."""
    import builtins
    import importlib
    import sys

    real_import = builtins.__import__

    def failing_import(name, globals=None, locals=None, fromlist=(), level=0):
        if name == "k3samd" and fromlist and "_C" in fromlist:
            raise ImportError("simulated missing extension")
        return real_import(name, globals, locals, fromlist, level)

    monkeypatch.setattr(builtins, "__import__", failing_import)
    sys.modules.pop("k3samd.ops", None)
    try:
        import k3samd.ops as ops2
        assert not ops2.native_available()
        with pytest.raises(RuntimeError, match="native extension"):
            ops2.stream_triad(torch.zeros(4), torch.zeros(4),
                              torch.zeros(4), 1.0)
    finally:
        monkeypatch.undo()
        sys.modules.pop("k3samd.ops", None)
        importlib.import_module("k3samd.ops")


def test_native_rejects_cpu_tensors():
    if not ops.native_available():
        pytest.skip("native extension not built")
    a = torch.zeros(16)
    with pytest.raises(Exception):
        ops.stream_triad(a, a.clone(), a.clone(), 1.0)


def test_mx_gemm16_argument_validation():
    """Binding-level checks fire before any launch (CPU-testable)."""
    import pytest
    import torch
    from k3samd import ops
    if not ops.native_available():
        pytest.skip("extension not built")
    a = torch.zeros(16, 128, dtype=torch.uint8)
    b = torch.zeros(16, 128, dtype=torch.uint8)
    with pytest.raises(RuntimeError, match="GPU"):
        ops.mx_gemm16(a, b, 0)  # CPU tensors rejected

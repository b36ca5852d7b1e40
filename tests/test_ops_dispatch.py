"""The op-dispatch policy (acco_amd/ops/__init__.py): CPU tensors take the
torch reference, CUDA tensors REQUIRE the HIP extension (loud RuntimeError
if it is not built — no silent eager fallback on a GPU box), and
ACCO_FORCE_REF=1 is the only override. Covers the driver's
"native code not loaded" failure mode from the CPU side."""

import types

import pytest
import torch

from acco_amd import ops


class _FakeCudaTensor:
    """Only `.is_cuda` is read by the dispatch predicate."""
    is_cuda = True


@pytest.fixture
def _ext_state():
    """Save/restore the module-level extension cache around each test."""
    saved = (ops._EXT, ops._EXT_ERR)
    yield
    ops._EXT, ops._EXT_ERR = saved


def test_cpu_tensor_uses_reference():
    assert ops._use_ref(torch.zeros(2), "rmsnorm_fwd") is True


def test_cuda_tensor_without_extension_raises(_ext_state, monkeypatch):
    monkeypatch.delenv("ACCO_FORCE_REF", raising=False)
    ops._EXT, ops._EXT_ERR = None, "forced-missing (test)"
    with pytest.raises(RuntimeError, match="not built"):
        ops._use_ref(_FakeCudaTensor(), "rmsnorm_fwd")
    with pytest.raises(RuntimeError, match="build_ext --inplace"):
        ops.hip_ext()


def test_force_ref_overrides(monkeypatch, _ext_state):
    monkeypatch.setenv("ACCO_FORCE_REF", "1")
    ops._EXT, ops._EXT_ERR = None, "forced-missing (test)"
    # no raise: the explicit A/B-debug override wins
    assert ops._use_ref(_FakeCudaTensor(), "rmsnorm_fwd") is True


def test_per_kernel_gap_falls_back_only_for_that_op(_ext_state, monkeypatch):
    """A built extension that lacks ONE kernel routes just that op to the
    reference (bring-up tolerance) while other ops stay native."""
    monkeypatch.delenv("ACCO_FORCE_REF", raising=False)
    fake_ext = types.SimpleNamespace(rmsnorm_fwd=lambda *a: None)
    ops._EXT, ops._EXT_ERR = fake_ext, None
    t = _FakeCudaTensor()
    assert ops._use_ref(t, "rmsnorm_fwd") is False      # present -> native
    assert ops._use_ref(t, "does_not_exist") is True    # absent -> reference


def test_ext_error_message_names_the_import_failure(_ext_state):
    ops._EXT, ops._EXT_ERR = None, "libmagic.so: cannot open"
    with pytest.raises(RuntimeError, match="libmagic"):
        ops.hip_ext()

"""CPU unit tests for the dispatch policies and small utilities added in
round 2 (no GPU needed)."""

import os

import pytest
import torch

from distributed_sigmoid_loss_amd import ops
from distributed_sigmoid_loss_amd.utils.profiling import HopStats


def test_save_g_policy_bounds(monkeypatch):
    monkeypatch.delenv("SIGLIP_SAVE_G", raising=False)
    # headline config fits comfortably
    assert ops.save_g_enabled(32768, 32768, "bf16")
    # beyond the kernel's 32-bit slab addressing: never
    assert not ops.save_g_enabled(65536, 65536, "bf16")
    monkeypatch.setenv("SIGLIP_SAVE_G", "1")
    assert not ops.save_g_enabled(65536, 65536, "bf16")   # still capped
    monkeypatch.setenv("SIGLIP_SAVE_G", "0")
    assert not ops.save_g_enabled(256, 256, "bf16")
    monkeypatch.delenv("SIGLIP_SAVE_G", raising=False)
    monkeypatch.setenv("SIGLIP_SAVE_G_MAX_BYTES", "1000000")
    assert not ops.save_g_enabled(32768, 32768, "bf16")   # over the cap
    assert ops.save_g_enabled(512, 512, "bf16")


def test_rowwise_policy_gate(monkeypatch):
    monkeypatch.delenv("SIGLIP_FP8_ROWWISE", raising=False)
    assert not ops.rowwise_ok(512, 512, 768)     # default off
    monkeypatch.setenv("SIGLIP_FP8_ROWWISE", "1")
    assert ops.rowwise_ok(512, 512, 768)
    assert not ops.rowwise_ok(500, 512, 768)     # misaligned b
    assert not ops.rowwise_ok(512, 500, 768)
    assert not ops.rowwise_ok(512, 512, 770)


def test_kernel_flags_defaults(monkeypatch):
    for k in ("SIGLIP_XCD_SWZ", "SIGLIP_GROUP_SWZ", "SIGLIP_NT_G",
              "SIGLIP_GROUP_M"):
        monkeypatch.delenv(k, raising=False)
    f = ops._kernel_flags()
    assert f & 1 and f & 2 and not (f & 4)
    assert (f >> 4) & 3 == 2          # GROUP_M=4 default for g-emitting
    f8 = ops._kernel_flags(default_gm="8")
    assert (f8 >> 4) & 3 == 0         # GROUP_M=8 for plain fwd
    monkeypatch.setenv("SIGLIP_NT_G", "1")
    assert ops._kernel_flags() & 4


def test_load_tuned_gemms_cpu_noop():
    # On a GPU-less host the loader must decline gracefully.
    if not torch.cuda.is_available():
        assert ops.load_tuned_gemms() is False
    # table ships with the package either way
    path = os.path.join(os.path.dirname(ops.__file__),
                        "tunableop_mi355x.csv")
    assert os.path.exists(path)


def test_hopstats_disabled_is_free():
    HopStats.set_enabled(False)
    with HopStats.record("x"):
        pass
    assert HopStats.summary() == {}


def test_reduce_out3_layout():
    buf = torch.zeros(8, 32)
    buf[3, 0] = 2.0
    buf[5, 0] = 1.0
    buf[0, 1] = 4.0
    buf[7, 2] = 8.0
    v = ops.reduce_out3(buf)
    assert v.tolist() == [3.0, 4.0, 8.0]


def test_banded_col_step_invariant(monkeypatch):
    """The banded column step must keep b*step*esz under 32-bit addressing,
    stay 256-aligned, and respect the SIGLIP_BANDED_STEP override."""
    monkeypatch.delenv("SIGLIP_BANDED_STEP", raising=False)
    for b in (131072, 262144, 1048576, 50000):
        step = ops.banded_col_step(b)
        assert step % 256 == 0 or step == 256
        if step > 256:
            assert b * step * 2 < 2 ** 32
        # one step wider would break the invariant (maximality)
        if (step // 256) * 256 == step and b * (step + 256) * 2 < 2 ** 32:
            assert False, f"step {step} not maximal for b={b}"
    monkeypatch.setenv("SIGLIP_BANDED_STEP", "512")
    assert ops.banded_col_step(131072) == 512
    monkeypatch.setenv("SIGLIP_BANDED_STEP", "7")   # floored to 256
    assert ops.banded_col_step(131072) == 256


def test_save_g_banded_policy_gates(monkeypatch):
    """Banded saved-g: bf16-only, honors the kill switches, and (without a
    GPU) reports unusable rather than crashing."""
    monkeypatch.delenv("SIGLIP_SAVE_G", raising=False)
    monkeypatch.delenv("SIGLIP_SAVE_G_BANDED", raising=False)
    assert not ops.save_g_banded_enabled(131072, 131072, "fp8")
    assert not ops.save_g_banded_enabled(131072, 131072, "mixed")
    monkeypatch.setenv("SIGLIP_SAVE_G", "0")
    assert not ops.save_g_banded_enabled(131072, 131072, "bf16")
    monkeypatch.delenv("SIGLIP_SAVE_G", raising=False)
    monkeypatch.setenv("SIGLIP_SAVE_G_BANDED", "0")
    assert not ops.save_g_banded_enabled(131072, 131072, "bf16")
    monkeypatch.delenv("SIGLIP_SAVE_G_BANDED", raising=False)
    if not torch.cuda.is_available():
        assert not ops.save_g_banded_enabled(131072, 131072, "bf16")

"""CPU regression tests for the NativeEngine capacity-split logic: a stub
core engine records the calls; the split boundaries and staging copies are
checked without a GPU (the numerics themselves are GPU-tested in
test_gpu_native_mp::test_oversized_collectives_split)."""

import pytest
import torch

from adapcc_amd.runtime import engine as eng_mod


class StubCore:
    DTYPE_F32, DTYPE_F16, DTYPE_BF16 = 0, 1, 2
    OP_SUM, OP_AVG, OP_MAX, OP_MIN = 0, 1, 2, 3


class StubEng:
    def __init__(self):
        self.calls = []

    def allreduce(self, ptr, numel, dt, op, active, average, stream):
        self.calls.append(("allreduce", numel))

    def reduce(self, ptr, numel, dt, op, root, active, stream):
        self.calls.append(("reduce", numel))

    def broadcast(self, ptr, numel, dt, root, stream):
        self.calls.append(("broadcast", numel))

    def all_gather(self, in_ptr, out_ptr, in_elems, dt, stream):
        self.calls.append(("all_gather", in_elems))

    def all_to_all(self, in_ptr, out_ptr, per, dt, stream):
        self.calls.append(("all_to_all", per))

    def reduce_scatter(self, in_ptr, out_ptr, out_elems, dt, op, active,
                       average, stream):
        self.calls.append(("reduce_scatter", out_elems))


def make_engine(world=4, cap_bytes=1024):
    e = eng_mod.NativeEngine.__new__(eng_mod.NativeEngine)
    e.rank, e.world_size, e.device = 0, world, 0
    e.cap_bytes = cap_bytes
    e._eng = StubEng()
    e._connected = True
    e._strategy_set = True
    return e


@pytest.fixture(autouse=True)
def stub_core(monkeypatch):
    monkeypatch.setattr(eng_mod, "_core", lambda: StubCore)
    monkeypatch.setattr(eng_mod.NativeEngine, "_stream",
                        staticmethod(lambda t: 0))
    # CPU tensors stand in for CUDA ones
    monkeypatch.setattr(
        eng_mod.NativeEngine, "_check",
        lambda self, t: None)


def test_allreduce_split_boundaries():
    e = make_engine(cap_bytes=1024)  # 256 f32 elements
    t = torch.zeros(1000)
    e.reduce(t)
    sizes = [n for (_, n) in e._eng.calls]
    assert sizes == [256, 256, 256, 232]


def test_all_gather_split_covers_all_columns():
    e = make_engine(world=4, cap_bytes=1024)  # per-rank cap 64 f32
    n = 150
    t = torch.arange(n, dtype=torch.float32)
    out = torch.zeros(4 * n)
    e.all_gather(out, t)
    calls = [c for c in e._eng.calls if c[0] == "all_gather"]
    assert [n_ for (_, n_) in calls] == [64, 64, 22]


def test_reduce_scatter_split():
    e = make_engine(world=4, cap_bytes=1024)
    out = torch.zeros(150)
    big = torch.zeros(600)
    e.reduce_scatter(out, big)
    calls = [c for c in e._eng.calls if c[0] == "reduce_scatter"]
    assert [n_ for (_, n_) in calls] == [64, 64, 22]


def test_all_to_all_split():
    e = make_engine(world=4, cap_bytes=1024)
    t = torch.zeros(4 * 100)
    out = torch.zeros_like(t)
    e.all_to_all(out, t)
    calls = [c for c in e._eng.calls if c[0] == "all_to_all"]
    assert [n_ for (_, n_) in calls] == [64, 36]

"""HostStager: CPU fallback, rotation bookkeeping, dtype mapping."""
import numpy as np
import torch

from agentcontrolplane_amd.engine.stage import HostStager


def test_cpu_fallback_values_and_dtypes():
    st = HostStager("cpu")
    t = st.tensor("a", [1, 2, 3], "int64")
    assert t.dtype == torch.int64 and t.tolist() == [1, 2, 3]
    f = st.fresh([[0.5, 1.5]], "float32")
    assert f.dtype == torch.float32 and f.shape == (1, 2)
    b = st.tensor("m", np.ones((2, 4), dtype=bool), "bool")
    assert b.dtype == torch.bool and b.all()


def test_returned_tensors_are_independent_snapshots_on_cpu():
    st = HostStager("cpu")
    src = [7, 8]
    t1 = st.tensor("x", src, "int64")
    src[0] = 99
    t2 = st.tensor("x", src, "int64")
    assert t1.tolist() == [7, 8] and t2.tolist() == [99, 8]


def test_rotation_counters():
    st = HostStager("cpu")
    for _ in range(7):
        st.step()
    assert st._slot == 7 % st.depth
    assert st._fresh_n == 0

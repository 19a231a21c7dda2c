"""Property-based fuzz of the pure-Python math paths (hypothesis).

Bounded example counts keep the suite fast; these cover input ranges
the fixed-case tests don't reach (odd image sizes, extreme epochs,
skewed weight distributions).
"""

import math

import pytest
import torch
from hypothesis import given, settings, strategies as st

from bdbnn_amd.data.loaders import _random_resized_crop_box
from bdbnn_amd.ops.kurtosis import KurtosisWeight
from bdbnn_amd.utils.utils import cpt_tk

FAST = settings(max_examples=25, deadline=None)


@FAST
@given(w=st.integers(8, 4096), h=st.integers(8, 4096),
       seed=st.integers(0, 2**31 - 1))
def test_crop_box_always_valid(w, h, seed):
    torch.manual_seed(seed)
    left, top, cw, ch = _random_resized_crop_box(w, h)
    assert 0 <= left and 0 <= top
    assert cw >= 1 and ch >= 1
    assert left + cw <= w and top + ch <= h


@FAST
@given(epoch=st.integers(0, 1000), tot=st.integers(1, 1000))
def test_cpt_tk_ranges(epoch, tot):
    # t spans 10^-2..10^1 over the schedule (ref EDE range,
    # utils/utils.py); k = max(1/t, 1) >= 1 always
    t, k = cpt_tk(min(epoch, tot), tot)
    t, k = float(t), float(k)
    assert 10.0 ** -2 - 1e-9 <= t <= 10.0 + 1e-6
    assert k >= 1.0 - 1e-9
    assert k == pytest.approx(max(1.0 / t, 1.0), rel=1e-6)


@FAST
@given(n=st.integers(8, 512), scale=st.floats(0.01, 100.0),
       shift=st.floats(-10.0, 10.0), target=st.floats(1.0, 3.0),
       seed=st.integers(0, 2**31 - 1))
def test_kurtosis_parity_formula(n, scale, shift, target, seed):
    # KurtosisWeight must equal the reference formula
    # mean(((w-mean)/std_unbiased)^4), (kurt-target)^2 for ANY scale and
    # shift (kurtosis is scale/shift invariant up to fp error)
    g = torch.Generator().manual_seed(seed)
    w = torch.randn(n, generator=g) * scale + shift
    kw = KurtosisWeight(w, "w", kurtosis_target=target)
    loss = kw.fn_regularization()
    mean, std = w.mean(), w.std()
    kurt = (((w - mean) / std) ** 4).mean()
    assert float(loss) == pytest.approx(float((kurt - target) ** 2),
                                        rel=1e-4, abs=1e-5)
    assert float(kw.kurtosis) == pytest.approx(float(kurt), rel=1e-4)

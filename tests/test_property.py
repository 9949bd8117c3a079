"""Property-based tests (hypothesis) — SURVEY §4.2 'hypothesis-generated
shapes'. CPU-only: forest layout invariants + resolver requirement algebra.
"""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from lambdipy_amd.shims.lpx_sklearn import ForestSoA
from lambdipy_amd.resolver.requirements import Requirement, parse_requirements
from tests.test_forest import soa_predict_proba_cpu


@settings(max_examples=20, deadline=None)
@given(trees=st.integers(1, 8), depth=st.integers(1, 6),
       feats=st.integers(2, 12), classes=st.integers(2, 6),
       samples=st.integers(1, 40), seed=st.integers(0, 2**31 - 1))
def test_heap_forest_cpu_reference_consistent(trees, depth, feats, classes,
                                              samples, seed):
    """Heap-layout forests: the CPU reference traversal yields per-class
    sums that (a) are rows of per-tree leaf distributions summed over
    `trees` trees, so each row sums to ~trees; (b) are deterministic."""
    f = ForestSoA.random(trees, depth, feats, classes, seed=seed, device=None)
    g = torch.Generator().manual_seed(seed ^ 0x5A5A)
    X = torch.randn(samples, feats, generator=g)
    out1 = soa_predict_proba_cpu(f, X)
    out2 = soa_predict_proba_cpu(f, X)
    assert np.allclose(out1, out2)
    assert out1.shape == (samples, classes)
    assert np.isfinite(out1).all()
    if f.layout == "heap":
        # heap-generated leaf rows are probability distributions and the
        # reference returns the across-tree MEAN -> rows sum to 1
        # fp16-representable leaf values: rows sum to 1 within ~5e-4
        assert np.allclose(out1.sum(axis=1), 1.0, rtol=1e-3)


@settings(max_examples=50, deadline=None)
@given(name=st.from_regex(r"[A-Za-z]([A-Za-z0-9_.\-]{0,18}[A-Za-z0-9])?", fullmatch=True),
       lo=st.integers(0, 20), hi=st.integers(21, 99))
def test_requirement_line_roundtrip(name, lo, hi):
    line = f"{name}>={lo},<{hi}"
    (r,) = parse_requirements(line)
    assert isinstance(r, Requirement)
    # canonicalized name: case/sep-insensitive identity
    (r2,) = parse_requirements(line.upper())
    assert r.name == r2.name
    from packaging.specifiers import SpecifierSet
    ss = SpecifierSet(r.specifier)
    assert ss.contains(str(lo)) or lo == 0 and not ss.contains("0")
    assert not ss.contains(str(hi))

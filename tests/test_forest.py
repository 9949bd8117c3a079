"""ForestSoA conversion correctness — validated on CPU against sklearn
(the GPU kernel is validated against this same SoA via a numpy traversal
reference; see test_ops_gpu.py for the on-device check)."""

import numpy as np
import pytest
import torch

from lambdipy_amd.shims.lpx_sklearn import ForestSoA


def soa_predict_proba_cpu(f: ForestSoA, X: np.ndarray) -> np.ndarray:
    """Reference traversal of the SoA arrays (mirrors the HIP kernels;
    handles both the classic [n,4] and packed depth-first [n,2] layouts)."""
    nodes = f.nodes.cpu().numpy()
    thr_f = np.ascontiguousarray(nodes[:, 1]).view(np.float32)
    leaves = f.leaf_values.cpu().numpy()
    roots = f.roots.cpu().numpy()
    S = X.shape[0]
    out = np.zeros((S, f.n_classes), dtype=np.float64)
    if f.layout == "heap":
        nint = (1 << f.depth) - 1
        feats = np.ascontiguousarray(nodes[:, 0]).view(np.uint32)
        for s in range(S):
            for tr in range(f.n_trees):
                nd = 0
                base = tr * nint
                for _ in range(f.depth):
                    feat = int(feats[base + nd])
                    nd = 2 * nd + 1 + (0 if X[s, feat] <= thr_f[base + nd] else 1)
                out[s] += leaves[tr * (1 << f.depth) + nd - nint]
        return (out / f.n_trees).astype(np.float32)
    if f.packed:
        meta = np.ascontiguousarray(nodes[:, 0]).view(np.uint32)
        for s in range(S):
            for r in roots:
                nd = int(r)
                while (meta[nd] >> 21) != 0x7FF:
                    feat = int(meta[nd] >> 21)
                    nd = nd + 1 if X[s, feat] <= thr_f[nd] else nd + int(meta[nd] & 0x1FFFFF)
                out[s] += leaves[nodes[nd, 1]]
    else:
        for s in range(S):
            for r in roots:
                nd = int(r)
                while nodes[nd, 0] >= 0:
                    feat = nodes[nd, 0]
                    nd = int(nodes[nd, 2]) if X[s, feat] <= thr_f[nd] else int(nodes[nd, 3])
                out[s] += leaves[nodes[nd, 2]]
    return (out / f.n_trees).astype(np.float32)


@pytest.fixture(scope="module")
def fitted_forest():
    sklearn = pytest.importorskip("sklearn")
    from sklearn.ensemble import RandomForestClassifier

    rng = np.random.default_rng(0)
    X = rng.standard_normal((400, 12)).astype(np.float32)
    y = (X[:, 0] + X[:, 3] * X[:, 5] > 0).astype(int) + (X[:, 2] > 1).astype(int)
    clf = RandomForestClassifier(n_estimators=12, max_depth=6, random_state=0)
    clf.fit(X, y)
    return clf, X


def test_from_sklearn_matches_predict_proba(fitted_forest):
    clf, X = fitted_forest
    f = ForestSoA.from_sklearn(clf, device=None)
    got = soa_predict_proba_cpu(f, X[:50])
    want = clf.predict_proba(X[:50]).astype(np.float32)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def test_from_sklearn_predict_classes(fitted_forest):
    clf, X = fitted_forest
    f = ForestSoA.from_sklearn(clf, device=None)
    proba = soa_predict_proba_cpu(f, X[:100])
    got = np.asarray(f.classes)[proba.argmax(axis=1)]
    want = clf.predict(X[:100])
    assert (got == want).mean() > 0.99  # ties may break differently


def test_random_forest_shapes():
    f = ForestSoA.random(n_trees=4, depth=3, n_features=8, n_classes=3,
                         seed=1, device=None)
    assert f.layout == "heap"  # complete forest -> heap layout
    assert f.nodes.shape == (4 * (2**3 - 1), 2)  # internal nodes only
    assert f.leaf_values.shape == (4 * 8, 3)
    assert f.depth == 3
    # leaf distributions normalized
    s = f.leaf_values.sum(dim=1)
    # leaf values are fp16-representable (exact compact kernel), so rows
    # sum to 1 within fp16 rounding (~5e-4)
    assert torch.allclose(s, torch.ones_like(s), atol=1e-3)


def test_random_forest_traversal_terminates():
    f = ForestSoA.random(n_trees=2, depth=4, n_features=5, n_classes=2,
                         seed=2, device=None)
    X = np.random.default_rng(3).standard_normal((16, 5)).astype(np.float32)
    proba = soa_predict_proba_cpu(f, X)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-3)


def test_regressor_conversion():
    sklearn = pytest.importorskip("sklearn")
    from sklearn.ensemble import RandomForestRegressor

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    y = X[:, 0] * 2 + np.sin(X[:, 1])
    reg = RandomForestRegressor(n_estimators=8, max_depth=5, random_state=0)
    reg.fit(X, y)
    f = ForestSoA.from_sklearn(reg, device=None)
    assert f.n_classes == 1
    got = soa_predict_proba_cpu(f, X[:40])[:, 0]
    want = reg.predict(X[:40]).astype(np.float32)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_pack_roundtrip_equivalence():
    """Packed traversal must agree exactly with the classic layout."""
    sklearn = pytest.importorskip("sklearn")
    from sklearn.ensemble import RandomForestClassifier

    rng = np.random.default_rng(2)
    X = rng.standard_normal((300, 9)).astype(np.float32)
    y = (X[:, 0] > 0).astype(int) + (X[:, 4] > 0.3)
    clf = RandomForestClassifier(n_estimators=9, max_depth=7, random_state=1)
    clf.fit(X, y)
    f = ForestSoA.from_sklearn(clf, device=None)
    assert f.packed
    got = soa_predict_proba_cpu(f, X[:80])
    want = clf.predict_proba(X[:80]).astype(np.float32)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def test_cpu_heap_reference_matches_naive():
    """The torch CPU heap reference (used by the gloo bench path) vs a
    per-sample python traversal."""
    import torch

    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    f = ForestSoA.random(n_trees=5, depth=4, n_features=6, n_classes=3,
                         seed=11, device="cpu")
    g = torch.Generator().manual_seed(12)
    X = torch.randn(17, 6, generator=g)
    proba = f.predict_proba_t(X)
    stride = (1 << 4) - 1
    nh = f.nodes.view(5, stride, 2)
    lv = f.leaf_values.view(5, 1 << 4, 3)
    for s in range(17):
        acc = torch.zeros(3)
        for t in range(5):
            nd = 0
            for _ in range(4):
                feat = int(nh[t, nd, 0])
                thr = nh[t, nd, 1].view(torch.float32).item()
                nd = 2 * nd + 1 + (0 if X[s, feat].item() <= thr else 1)
            acc += lv[t, nd - stride]
        assert torch.allclose(proba[s], acc / 5, atol=1e-6), f"sample {s}"


@pytest.mark.gpu
def test_gpu_heap_kernel_matches_cpu_reference():
    """HIP heap kernel vs the plain-torch CPU reference on the same forest."""
    import torch

    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    f = ForestSoA.random(n_trees=32, depth=8, n_features=16, n_classes=8,
                         seed=3, device="cuda")
    g = torch.Generator().manual_seed(4)
    X = torch.randn(4096, 16, generator=g)
    proba_gpu = f.predict_proba_t(X.cuda()).cpu()
    fc = ForestSoA.random(n_trees=32, depth=8, n_features=16, n_classes=8,
                          seed=3, device="cpu")
    proba_cpu = fc.predict_proba_t(X)
    assert torch.allclose(proba_gpu, proba_cpu, atol=1e-5)

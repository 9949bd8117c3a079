"""GPU numerics tests: every gfx950 kernel vs a plain PyTorch fp32 reference
(asymmetric operands — transpose-detecting, guide §5.4 rule 16) plus
determinism double-runs. Run on the MI355X box: pytest tests -m gpu."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda"


@pytest.fixture(scope="module")
def ops():
    from lambdipy_amd import ops as _ops

    # fail-loud contract: on a GPU box the extension must be present
    assert _ops.extension_available(), f"extension missing: {_ops.extension_error()}"
    return _ops


def _rand(shape, dtype=torch.float32, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return torch.randn(*shape, generator=g, dtype=torch.float32).to(DEV).to(dtype)


# ---- GEMM ----------------------------------------------------------------

@pytest.mark.parametrize("m,n,k", [(256, 256, 256), (128, 128, 64),
                                   (512, 384, 320), (100, 130, 70),
                                   (4096, 4096, 4096)])
def test_gemm_bf16_vs_torch(ops, m, n, k):
    A = _rand((m, k), torch.bfloat16, seed=m + n)
    Bt = _rand((n, k), torch.bfloat16, seed=m * 3 + k)
    C = ops.gemm_bf16(A, Bt)
    ref = (A.float() @ Bt.float().t())
    err = (C - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 2e-2, f"rel err {err/scale}"


def test_gemm_bf16_out_dtype(ops):
    A = _rand((128, 128), torch.bfloat16)
    Bt = _rand((128, 128), torch.bfloat16, seed=9)
    C = ops.gemm_bf16(A, Bt, bf16_out=True)
    assert C.dtype == torch.bfloat16
    ref = (A.float() @ Bt.float().t())
    assert (C.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-6) < 3e-2


def test_gemm_deterministic(ops):
    A = _rand((512, 512), torch.bfloat16, seed=4)
    Bt = _rand((512, 512), torch.bfloat16, seed=5)
    c1 = ops.gemm_bf16(A, Bt)
    c2 = ops.gemm_bf16(A, Bt)
    assert torch.equal(c1, c2)


# ---- GEMV ----------------------------------------------------------------

@pytest.mark.parametrize("b,m,k", [(1, 256, 1024), (8, 512, 4096), (3, 100, 77)])
def test_gemv_bf16(ops, b, m, k):
    A = _rand((b, m, k), torch.bfloat16, seed=b)
    x = _rand((b, k), torch.bfloat16, seed=b + 1)
    y = ops.gemv(A, x)
    ref = torch.einsum("bmk,bk->bm", A.float(), x.float())
    err = (y - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 2e-2, err


def test_gemv_f32_2d(ops):
    A = _rand((300, 500), torch.float32, seed=7)
    x = _rand((500,), torch.float32, seed=8)
    y = ops.gemv(A, x)
    ref = A @ x
    assert torch.allclose(y, ref, rtol=1e-4, atol=1e-4)


# ---- SpMV ----------------------------------------------------------------

def _random_csr(rows, cols, nnz_per_row, seed=0):
    rng = np.random.default_rng(seed)
    indices = np.sort(
        rng.integers(0, cols, (rows, nnz_per_row)).astype(np.int32), axis=1
    ).reshape(-1)
    indptr = np.arange(0, (rows + 1) * nnz_per_row, nnz_per_row, dtype=np.int32)
    data = rng.standard_normal(rows * nnz_per_row).astype(np.float32)
    return indptr, indices, data


@pytest.mark.parametrize("rows,cols,nnz", [(1000, 800, 7), (512, 512, 100),
                                           (3000, 3000, 33)])
def test_spmv_vs_dense(ops, rows, cols, nnz):
    indptr, indices, data = _random_csr(rows, cols, nnz, seed=rows)
    x = np.random.default_rng(1).standard_normal(cols).astype(np.float32)
    ti = torch.from_numpy(indptr).to(DEV)
    tj = torch.from_numpy(indices).to(DEV)
    td = torch.from_numpy(data).to(DEV)
    tx = torch.from_numpy(x).to(DEV)
    y = ops.spmv_csr(ti, tj, td, tx).cpu().numpy()
    # dense reference
    dense = np.zeros((rows, cols), dtype=np.float64)
    for r in range(rows):
        for j in range(indptr[r], indptr[r + 1]):
            dense[r, indices[j]] += data[j]
    ref = dense @ x.astype(np.float64)
    np.testing.assert_allclose(y, ref, rtol=1e-4, atol=1e-4)


# ---- FFT -----------------------------------------------------------------

@pytest.mark.parametrize("batch,n", [(4, 256), (16, 1024), (2, 4096),
                                     (3, 8192), (128, 64), (1, 2)])
def test_fft_vs_torch(ops, batch, n):
    g = torch.Generator(device="cpu").manual_seed(n)
    x = (torch.randn(batch, n, generator=g) +
         1j * torch.randn(batch, n, generator=g)).to(torch.complex64).to(DEV)
    got = ops.fft(x)
    ref = torch.fft.fft(x.to(torch.complex128))
    err = (got.to(torch.complex128) - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err / scale < 1e-4, err / scale


def test_ifft_roundtrip(ops):
    g = torch.Generator(device="cpu").manual_seed(0)
    x = (torch.randn(8, 512, generator=g) +
         1j * torch.randn(8, 512, generator=g)).to(torch.complex64).to(DEV)
    rt = ops.ifft(ops.fft(x))
    assert (rt - x).abs().max().item() < 1e-3


# ---- pairwise ------------------------------------------------------------

@pytest.mark.parametrize("m,n,k", [(128, 128, 32), (500, 300, 64), (64, 64, 7),
                                   (1000, 1000, 128)])
def test_pairwise_vs_cdist(ops, m, n, k):
    X = _rand((m, k), seed=m)
    Y = _rand((n, k), seed=n + 1)
    D = ops.pairwise_sqeuclidean(X, Y)
    ref = torch.cdist(X.double(), Y.double()) ** 2
    err = (D.double() - ref).abs().max().item()
    assert err < 1e-2 * (ref.max().item() + 1), err


@pytest.mark.parametrize("variant,rel_bound", [("5", 3e-4), ("6", 1e-6),
                                               ("7", 3e-4), ("8", 1e-6),
                                               ("9", 3e-4)])
def test_pairwise_bf16_split_variants(ops, monkeypatch, variant, rel_bound):
    """v5/v6 bf16-split emulation: x=hi+mid(+lo) exact truncation splits,
    4/6 cross products on the bf16 MFMA pipe.  v6 keeps hi*lo terms ->
    f32-grade (~2^-24); v5 drops them -> ~2^-16/term, still far inside the
    sklearn pairwise tolerance.  Checked vs a float64 reference + both must
    be deterministic."""
    m, n, k = 1280, 1100, 200  # non-multiples of the 256/32 tile dims
    X = _rand((m, k), seed=5)
    Y = _rand((n, k), seed=6)
    monkeypatch.setenv("LPX_PW_V2", variant)
    D = ops.pairwise_sqeuclidean(X, Y)
    D2 = ops.pairwise_sqeuclidean(X, Y)
    assert torch.equal(D, D2), "nondeterministic split-pairwise"
    ref = torch.cdist(X.double(), Y.double()) ** 2
    err = (D.double() - ref).abs().max().item()
    assert err < rel_bound * (ref.max().item() + 1), \
        f"v{variant} err {err:.3e} bound {rel_bound * (ref.max().item() + 1):.3e}"


# ---- RandomForest --------------------------------------------------------

def test_rf_vs_cpu_reference(ops):
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    from tests.test_forest import soa_predict_proba_cpu

    f = ForestSoA.random(n_trees=31, depth=8, n_features=20, n_classes=3,
                         seed=11, device=None)
    rng = np.random.default_rng(5)
    X = rng.standard_normal((500, 20)).astype(np.float32)
    ref = soa_predict_proba_cpu(f, X)
    fg = f.to(DEV)
    got = fg.predict_proba_t(torch.from_numpy(X).to(DEV)).cpu().numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-5)


def test_rf_vs_sklearn(ops):
    sklearn = pytest.importorskip("sklearn")
    from sklearn.ensemble import RandomForestClassifier
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    rng = np.random.default_rng(0)
    X = rng.standard_normal((600, 16)).astype(np.float32)
    y = ((X[:, 0] > 0) & (X[:, 2] < 1)).astype(int) + (X[:, 5] > 0.5)
    clf = RandomForestClassifier(n_estimators=20, max_depth=8, random_state=0)
    clf.fit(X, y)
    f = ForestSoA.from_sklearn(clf, device=DEV)
    got = f.predict_proba(X)
    want = clf.predict_proba(X).astype(np.float32)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_rf_many_classes_chunking(ops):
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    from tests.test_forest import soa_predict_proba_cpu

    f = ForestSoA.random(n_trees=6, depth=5, n_features=10, n_classes=11,
                         seed=3, device=None)
    rng = np.random.default_rng(6)
    X = rng.standard_normal((100, 10)).astype(np.float32)
    ref = soa_predict_proba_cpu(f, X)
    got = f.to(DEV).predict_proba_t(torch.from_numpy(X).to(DEV)).cpu().numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-5)


def test_rf_wide_features_no_lds_path(ops):
    """F > 63 exercises the non-staged (global X) kernel variant."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    from tests.test_forest import soa_predict_proba_cpu

    f = ForestSoA.random(n_trees=5, depth=6, n_features=100, n_classes=2,
                         seed=9, device=None)
    rng = np.random.default_rng(10)
    X = rng.standard_normal((200, 100)).astype(np.float32)
    ref = soa_predict_proba_cpu(f, X)
    got = f.to(DEV).predict_proba_t(torch.from_numpy(X).to(DEV)).cpu().numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-5)


def test_rf_deterministic(ops):
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    f = ForestSoA.random(n_trees=64, depth=10, n_features=32, n_classes=8,
                         seed=1, device=DEV)
    X = _rand((4096, 32))
    p1 = f.predict_proba_t(X)
    p2 = f.predict_proba_t(X)
    assert torch.equal(p1, p2)


# ---- fp8 (config 5 path) -------------------------------------------------

def test_quantize_fp8_matches_torch_e4m3(ops):
    g = torch.Generator(device="cpu").manual_seed(0)
    x = (torch.randn(4096, generator=g) * 100).to(DEV)
    got = ops.ext().quantize_fp8(x, 1.0)
    ref = x.to(torch.float8_e4m3fn).view(torch.uint8)
    # allow the rare off-by-one at exact rounding ties
    mism = (got != ref).float().mean().item()
    assert mism < 1e-3, f"{mism*100:.3f}% mismatch vs torch e4m3fn"


def test_gemm_fp8_vs_dequant_ref(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(1)
    M, N, K = 256, 192, 128
    A = torch.randn(M, K, generator=g).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    sa, sb = 0.25, 0.5
    Aq = e.quantize_fp8(A, sa)
    Btq = e.quantize_fp8(Bt, sb)
    C = e.gemm_fp8(Aq, Btq, sa * sb)
    # reference: dequantized operands, f32 matmul
    Ad = Aq.view(torch.float8_e4m3fn).float() * sa
    Btd = Btq.view(torch.float8_e4m3fn).float() * sb
    ref = Ad @ Btd.t()
    err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err


def _mx_scales(sc, R, groups):
    """Tile-major [KT, R, 4] E8M0 bytes -> [R, groups, 2] scale factors."""
    s = sc.permute(1, 0, 2).reshape(R, -1, 2, 2)   # [R, KT, ghalf, kb]
    s = s.reshape(R, s.shape[1] * 2, 2)[:, :groups]
    return torch.pow(2.0, s.float() - 127.0)


def _dequant_mx(q, sc):
    """Reconstruct f32 from MX-quantized (shuffled) bytes + E8M0 scales.

    Shuffled 64-group order is [q0|q2|q1|q3]; scale kb covers the contiguous
    original block {q_2kb, q_2kb+1} = shuffled slots {kb, kb+2}."""
    R, K = q.shape
    g = K // 64
    d = q.view(R, g, 4, 16).view(torch.float8_e4m3fn).float()
    scales = _mx_scales(sc, R, g)                        # [R, g, 2]
    d = d * scales[:, :, [0, 1, 0, 1]].unsqueeze(-1)     # slots q0,q2,q1,q3
    return d[:, :, [0, 2, 1, 3], :].reshape(R, K)        # unshuffle


def test_quantize_mx_roundtrip(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(7)
    # wide dynamic range across blocks: per-row magnitude ramp
    x = (torch.randn(64, 256, generator=g) *
         torch.logspace(-6, 6, 64, base=2.0).unsqueeze(1)).to(DEV)
    q, sc = e.quantize_mx(x)
    xd = _dequant_mx(q, sc)
    # per-element error bounded by the e4m3 step within each contiguous
    # 32-element MX block (relative to the block amax)
    err = (xd - x).abs().view(64, 4, 2, 32)
    blk_amax = x.view(64, 4, 2, 32).abs().amax(dim=-1, keepdim=True)
    err_blk = err / blk_amax.clamp(min=1e-30)
    assert err_blk.max().item() < 0.08, err_blk.max().item()


def test_gemm_mx_vs_dequant_ref(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(8)
    M, N, K = 512, 512, 256
    A = (torch.randn(M, K, generator=g) * 3).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    qa, sa = e.quantize_mx(A)
    qb, sb = e.quantize_mx(Bt)
    C = e.gemm_mx(qa, sa, qb, sb)
    ref = _dequant_mx(qa, sa) @ _dequant_mx(qb, sb).t()
    err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err
    # end-to-end quantization error vs exact f32 matmul: ~4% Frobenius is
    # the expected e4m3 noise floor for random data (element RMS ~3%)
    exact = A @ Bt.t()
    rel = (C - exact).norm().item() / exact.norm().item()
    assert rel < 0.06, rel


def test_gemm_mx_tail_shapes(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(9)
    M, N, K = 300, 200, 192  # non-multiples of the 256x256 tile
    A = torch.randn(M, K, generator=g).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    qa, sa = e.quantize_mx(A)
    qb, sb = e.quantize_mx(Bt)
    C = e.gemm_mx(qa, sa, qb, sb)
    ref = _dequant_mx(qa, sa) @ _dequant_mx(qb, sb).t()
    err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err


_E2M1 = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
         -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0]


def _dequant_mx4(q, sc, K):
    """Reconstruct f32 from MX-fp4 nibbles + E8M0 scales (low nibble first;
    LINEAR layout — fp4 lanes own their whole 32-element block, so no
    quarter shuffle; see scripts/fp4_diag.py)."""
    R = q.shape[0]
    g = K // 64
    lut = torch.tensor(_E2M1, device=q.device)
    lo = lut[(q & 0xF).long()]
    hi = lut[(q >> 4).long()]
    d = torch.stack([lo, hi], dim=-1).reshape(R, g, 2, 32)
    scales = _mx_scales(sc, R, g)
    return (d * scales.unsqueeze(-1)).reshape(R, K)


def test_quantize_mx_fp4_roundtrip(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(12)
    x = (torch.randn(64, 256, generator=g) *
         torch.logspace(-6, 6, 64, base=2.0).unsqueeze(1)).to(DEV)
    q, sc = e.quantize_mx4(x)
    xd = _dequant_mx4(q, sc, 256)
    err = (xd - x).abs().view(64, 4, 2, 32)
    blk_amax = x.view(64, 4, 2, 32).abs().amax(dim=-1, keepdim=True)
    # e2m1 worst step/amax: gap 2 around 4..6, amax in (3,6] -> <= 1/3
    assert (err / blk_amax.clamp(min=1e-30)).max().item() < 0.35


def test_gemm_mx_fp4_vs_dequant_ref(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(13)
    M, N, K = 512, 512, 384
    A = (torch.randn(M, K, generator=g) * 2).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    qa, sa = e.quantize_mx4(A)
    qb, sb = e.quantize_mx4(Bt)
    C = e.gemm_mx4(qa, sa, qb, sb)
    ref = _dequant_mx4(qa, sa, K) @ _dequant_mx4(qb, sb, K).t()
    err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err
    # fp4 quantization noise vs exact matmul (e2m1 element RMS ~10%)
    exact = A @ Bt.t()
    rel = (C - exact).norm().item() / exact.norm().item()
    assert rel < 0.2, rel


def test_gemm_mx_w4_vs_dequant_ref(ops):
    # mixed: fp8-MX activations x fp4-MX weights
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(19)
    M, N, K = 512, 512, 384
    A = (torch.randn(M, K, generator=g) * 2).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    qa, sa = e.quantize_mx(A)
    qb, sb = e.quantize_mx4(Bt)
    C = e.gemm_mx_w4(qa, sa, qb, sb)
    ref = _dequant_mx(qa, sa) @ _dequant_mx4(qb, sb, K).t()
    err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err
    # vs exact f32: activations e4m3 + weights e2m1 -> ~e2m1-dominated noise
    exact = A @ Bt.t()
    rel = (C - exact).norm().item() / exact.norm().item()
    assert rel < 0.15, rel


def test_gemm_fp8_tall_skinny(ops):
    # N<=64, large M (ResNet layer1 1x1 conv shape class, K-tail 147)
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(22)
    for M, N, K in ((4096, 64, 147), (2048, 48, 64), (2304, 64, 128)):
        A = torch.randn(M, K, generator=g).to(DEV)
        Bt = torch.randn(N, K, generator=g).to(DEV)
        bias = torch.randn(N, generator=g).to(DEV)
        Aq = e.quantize_fp8(A, 1.0)
        Btq = e.quantize_fp8(Bt, 1.0)
        C = e.gemm_fp8(Aq, Btq, 1.0, bias, True)
        ref = torch.relu(Aq.view(torch.float8_e4m3fn).float()
                         @ Btq.view(torch.float8_e4m3fn).float().t() + bias)
        err = (C - ref).abs().max().item()
        assert err < 1e-2, (M, N, K, err)


def test_gemm_fp8_routing_tiers(ops, monkeypatch):
    """Grid-fill-aware dispatch (scripts/ab_fp8_route.py): tall-skinny now
    serves N<=256 by default, and LPX_FP8_ROUTE can force the M-split
    (8phx on aligned rows + nt tail) and nt256 tiers — every route must
    agree with the dequantized-operand reference on the same shape."""
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(31)

    def check(M, N, K, route):
        if route is None:
            monkeypatch.delenv("LPX_FP8_ROUTE", raising=False)
        else:
            monkeypatch.setenv("LPX_FP8_ROUTE", route)
        A = torch.randn(M, K, generator=g).to(DEV)
        Bt = torch.randn(N, K, generator=g).to(DEV)
        Aq = e.quantize_fp8(A, 1.0)
        Btq = e.quantize_fp8(Bt, 1.0)
        C = e.gemm_fp8(Aq, Btq, 1.0, None, False)
        ref = (Aq.view(torch.float8_e4m3fn).float()
               @ Btq.view(torch.float8_e4m3fn).float().t())
        err = (C - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
        assert err < 1e-3, (M, N, K, route, err)

    check(2048, 128, 320, None)   # widened tall-skinny default (N=128)
    check(2304, 256, 512, None)   # tall-skinny default (N=256)
    check(832, 512, 128, "s")     # forced M-split: 8phx 768 rows + 64-row nt tail
    check(832, 512, 96, "b")      # forced nt256
    check(3136, 512, 96, None)    # under-filled grid -> nt
    monkeypatch.delenv("LPX_FP8_ROUTE", raising=False)


def test_linear_w4_matches_f32(ops):
    from lambdipy_amd.shims.lpx_torch import LinearW4
    g = torch.Generator(device="cpu").manual_seed(20)
    M, IN, OUT = 256, 256, 256
    w = (torch.randn(OUT, IN, generator=g) * 0.1).to(DEV)
    b = torch.randn(OUT, generator=g).to(DEV)
    x = torch.randn(M, IN, generator=g).to(DEV)
    lin = LinearW4(w, b)
    y = lin(x)
    ref = x @ w.t() + b
    rel = (y - ref).norm().item() / ref.norm().item()
    assert rel < 0.15, rel  # e2m1 weights dominate the quantization noise


def test_gemm_fp8_scaled_mfma_path(ops):
    # 256-multiple shapes dispatch to the unit-scale scaled-MFMA kernel
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(11)
    M, N, K = 512, 512, 384
    A = torch.randn(M, K, generator=g).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    bias = torch.randn(N, generator=g).to(DEV)
    sa, sb = 0.25, 0.5
    Aq = e.quantize_fp8(A, sa)
    Btq = e.quantize_fp8(Bt, sb)
    Ad = Aq.view(torch.float8_e4m3fn).float() * sa
    Btd = Btq.view(torch.float8_e4m3fn).float() * sb
    C = e.gemm_fp8(Aq, Btq, sa * sb)
    ref = Ad @ Btd.t()
    assert (C - ref).abs().max().item() / ref.abs().max().item() < 1e-3
    C2 = e.gemm_fp8(Aq, Btq, sa * sb, bias, True)
    ref2 = torch.relu(ref + bias)
    assert (C2 - ref2).abs().max().item() < 1e-2


def test_determinism_double_run(ops):
    """SURVEY §4.2 race check: every kernel is bitwise deterministic across
    back-to-back runs on identical inputs (no atomics/race in the hot path
    except the reduction kernels, which are unordered by design and tested
    for tolerance elsewhere)."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(21)
    a = torch.randn(512, 512, generator=g).to(DEV).to(torch.bfloat16)
    assert torch.equal(ops.gemm_bf16(a, a), ops.gemm_bf16(a, a))
    aq = e.quantize_fp8(a.float(), 1.0)
    assert torch.equal(e.gemm_fp8(aq, aq, 1.0), e.gemm_fp8(aq, aq, 1.0))
    qa, sa = e.quantize_mx(a.float().contiguous())
    assert torch.equal(e.gemm_mx(qa, sa, qa, sa), e.gemm_mx(qa, sa, qa, sa))
    f = ForestSoA.random(32, 8, 16, 4, seed=3, device=DEV)
    X = torch.randn(10000, 16, generator=g).to(DEV)
    assert torch.equal(f.predict_proba_t(X), f.predict_proba_t(X))


def test_conv_fp8_8ph_vs_unfold_ref(ops):
    # shape chosen to hit the 8-phase glds conv (M%256, KO%256, K%128)
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(17)
    N, H, W, C, KO = 4, 16, 16, 128, 256
    x = torch.randn(N, H, W, C, generator=g).to(DEV)
    w = torch.randn(KO, 3 * 3 * C, generator=g).to(DEV)
    xq = e.quantize_fp8(x, 1.0)
    wq = e.quantize_fp8(w, 1.0)
    got = e.conv_fp8(xq, wq, H, W, C, 3, 3, 1, 1, 1.0)
    xd = xq.view(torch.float8_e4m3fn).float()
    wd = wq.view(torch.float8_e4m3fn).float()
    # reference: unfold on NCHW then matmul
    unf = torch.nn.functional.unfold(
        xd.permute(0, 3, 1, 2), 3, padding=1)  # [N, C*3*3, L] (C-major taps)
    # our patch layout is (r, s, c) row-major with c fastest: rearrange
    unf = unf.view(N, C, 3, 3, -1).permute(0, 4, 2, 3, 1).reshape(-1, 3 * 3 * C)
    ref = unf @ wd.t()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    assert err < 1e-3, err


def test_resnet_graph_capture_matches_eager(ops):
    from lambdipy_amd.shims.lpx_torch import ResNet50FP8
    m = ResNet50FP8(device=DEV, seed=0)
    x = torch.randn(8, 3, 224, 224, device=DEV)
    m(x)  # calibration
    eager = m(x).clone()
    m.capture_graph(8)
    replay = m(x).clone()
    rel = (replay - eager).norm().item() / (eager.norm().item() + 1e-9)
    assert rel < 1e-4, rel


def test_maxpool_nhwc_matches_torch(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(16)
    x = torch.randn(2, 13, 13, 32, generator=g).to(DEV)
    got = e.maxpool2d_nhwc(x, 3, 2, 1)
    import torch.nn.functional as F
    ref = F.max_pool2d(x.permute(0, 3, 1, 2), 3, 2, 1).permute(0, 2, 3, 1)
    assert torch.equal(got, ref)


def test_gemm_fp8_q8_matches_two_step(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(14)
    for M, N, K in ((512, 512, 256), (300, 200, 192)):  # fused + fallback
        A = torch.randn(M, K, generator=g).to(DEV)
        Bt = torch.randn(N, K, generator=g).to(DEV)
        Aq = e.quantize_fp8(A, 1.0)
        Btq = e.quantize_fp8(Bt, 1.0)
        out_scale = 0.37
        q8 = e.gemm_fp8_q8(Aq, Btq, 1.0, None, True, out_scale)
        ref = e.quantize_fp8(e.gemm_fp8(Aq, Btq, 1.0, None, True), out_scale)
        mism = (q8 != ref).float().mean().item()
        assert mism < 2e-3, f"{M}x{N}x{K}: {mism*100:.3f}% byte mismatch"


def test_conv_fp8_q8_matches_two_step(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(15)
    N, H, W, C, KO = 2, 14, 14, 32, 64
    x = torch.randn(N, H, W, C, generator=g).to(DEV)
    w = torch.randn(KO, 3 * 3 * C, generator=g).to(DEV)
    xq = e.quantize_fp8(x, 1.0)
    wq = e.quantize_fp8(w, 1.0)
    out_scale = 0.5
    q8 = e.conv_fp8_q8(xq, wq, H, W, C, 3, 3, 1, 1, 1.0, None, True, out_scale)
    f32 = e.conv_fp8(xq, wq, H, W, C, 3, 3, 1, 1, 1.0, None, True)
    ref = e.quantize_fp8(f32, out_scale)
    mism = (q8 != ref.view_as(q8)).float().mean().item()
    assert mism < 2e-3, f"{mism*100:.3f}% byte mismatch"


def test_resnet_bottleneck_fused_consistency(ops):
    from lambdipy_amd.shims.lpx_torch import Bottleneck
    g = torch.Generator().manual_seed(3)
    blk = Bottleneck(64, 16, 64, 1, DEV, g)
    x = torch.randn(2, 8, 8, 64, generator=torch.Generator().manual_seed(4)).to(DEV)
    y_cal = blk(x)        # calibration pass (unfused)
    y_fused = blk(x)      # fused chain, same input
    rel = (y_fused - y_cal).norm().item() / (y_cal.norm().item() + 1e-9)
    assert rel < 0.05, rel


def test_gemm_fp8_bias_relu(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(2)
    M, N, K = 128, 128, 64
    A = torch.randn(M, K, generator=g).to(DEV)
    Bt = torch.randn(N, K, generator=g).to(DEV)
    bias = torch.randn(N, generator=g).to(DEV)
    Aq = e.quantize_fp8(A, 1.0)
    Btq = e.quantize_fp8(Bt, 1.0)
    C = e.gemm_fp8(Aq, Btq, 1.0, bias, True)
    Ad = Aq.view(torch.float8_e4m3fn).float()
    Btd = Btq.view(torch.float8_e4m3fn).float()
    ref = torch.relu(Ad @ Btd.t() + bias)
    assert (C - ref).abs().max().item() < 1e-2


def test_im2col_fp8_matches_unfold(ops):
    e = ops.ext()
    g = torch.Generator(device="cpu").manual_seed(3)
    N, H, W, C = 2, 14, 14, 8
    R = S = 3
    stride, pad = 2, 1
    x = torch.randn(N, H, W, C, generator=g).to(DEV)
    got = e.im2col_fp8(x, R, S, stride, pad, 1.0).view(torch.float8_e4m3fn).float()
    # reference via torch.nn.functional.unfold on NCHW
    import torch.nn.functional as TF
    xn = x.permute(0, 3, 1, 2).contiguous()  # NCHW
    ref = TF.unfold(xn, (R, S), padding=pad, stride=stride)  # [N, C*R*S, L]
    HO = (H + 2 * pad - R) // stride + 1
    WO = (W + 2 * pad - S) // stride + 1
    # unfold col order is (c, r, s); ours is (r, s, c)
    ref = ref.view(N, C, R, S, HO * WO).permute(0, 4, 2, 3, 1).reshape(N * HO * WO, R * S * C)
    refq = ref.to(torch.float8_e4m3fn).float()
    assert (got - refq).abs().max().item() < 1e-6


def test_resnet50_fp8_forward(ops):
    from lambdipy_amd.shims.lpx_torch import ResNet50FP8

    model = ResNet50FP8(n_classes=100, device=DEV, seed=1)
    x = torch.randn(2, 3, 64, 64, generator=torch.Generator().manual_seed(0)).to(DEV)
    y1 = model(x)
    assert y1.shape == (2, 100)
    assert torch.isfinite(y1).all()
    y2 = model(x)
    assert torch.equal(y1, y2)  # deterministic inference


def test_rf_heap_vs_cpu_reference(ops):
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA
    from tests.test_forest import soa_predict_proba_cpu

    f = ForestSoA.random(n_trees=17, depth=7, n_features=12, n_classes=5,
                         seed=21, device=None)
    assert f.layout == "heap"
    rng = np.random.default_rng(8)
    X = rng.standard_normal((300, 12)).astype(np.float32)
    ref = soa_predict_proba_cpu(f, X)
    got = f.to(DEV).predict_proba_t(torch.from_numpy(X).to(DEV)).cpu().numpy()
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-5)


def test_rf_heap_deterministic_and_silp(ops):
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    f = ForestSoA.random(n_trees=64, depth=9, n_features=32, n_classes=8,
                         seed=2, device=DEV)
    X = _rand((65536, 32))
    e = ops.ext()
    p1 = e.rf_predict_heap(f.nodes, f.leaf_values, f.depth, X, 1)
    p2 = e.rf_predict_heap(f.nodes, f.leaf_values, f.depth, X, 2)
    p4 = e.rf_predict_heap(f.nodes, f.leaf_values, f.depth, X, 4)
    assert torch.equal(p1, p2) and torch.equal(p1, p4)


def _heap_pairs_roots(nodes, depth):
    """Repack flat heap nodes [T*stride, 2] into the v6 child-pair layout."""
    stride = (1 << depth) - 1
    T = nodes.shape[0] // stride
    nh = nodes.view(T, stride, 2)
    ppt = (1 << (depth - 1)) - 1
    idx = torch.arange(ppt, device=nodes.device)
    pairs = (torch.cat([nh[:, 2 * idx + 1, :], nh[:, 2 * idx + 2, :]], dim=2)
             .reshape(T * ppt, 4).contiguous())
    return pairs, nh[:, 0, :].contiguous()


def test_rf_heap_pair_matches_v4(ops):
    """v6 child-pair layout (the routed default) against the flat v4 kernel."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    e = ops.ext()
    for depth in (5, 10):  # odd and even depths
        f = ForestSoA.random(n_trees=32, depth=depth, n_features=16,
                             n_classes=6, seed=4, device=DEV)
        X = _rand((32768, 16))
        ref = e.rf_predict_heap(f.nodes, f.leaf_values, depth, X, 1, -2)
        pairs, roots = _heap_pairs_roots(f.nodes, depth)
        got = e.rf_predict_heap_pair(pairs, roots, f.leaf_values, depth, X, 1)
        assert torch.allclose(got, ref, atol=1e-5), (got - ref).abs().max()


def test_rf_heap_quad_matches_v4(ops):
    """v7 quad-record layout (A/B only — slower, kept for evidence)."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    e = ops.ext()
    depth = 8
    f = ForestSoA.random(n_trees=16, depth=depth, n_features=12, n_classes=4,
                         seed=5, device=DEV)
    X = _rand((16384, 12))
    ref = e.rf_predict_heap(f.nodes, f.leaf_values, depth, X, 1, -2)
    stride = (1 << depth) - 1
    T = f.nodes.shape[0] // stride
    nh = f.nodes.view(T, stride, 2)
    qpt = ((1 << depth) - 1) // 3
    quads = torch.zeros(T, qpt, 16, dtype=torch.int32, device=DEV)
    for k in range(depth // 2):
        base = (4 ** k - 1) // 3
        o = torch.arange(4 ** k, device=DEV)
        nn = (1 << (2 * k)) - 1 + o
        quads[:, base + o, 0:2] = nh[:, 2 * nn + 1, :]
        quads[:, base + o, 2:4] = nh[:, 2 * nn + 2, :]
        if 2 * k + 2 < depth:
            for j in range(4):
                quads[:, base + o, 4 + 2 * j:6 + 2 * j] = nh[:, 4 * nn + 3 + j, :]
    quads = quads.reshape(T * qpt, 16).contiguous()
    roots = nh[:, 0, :].contiguous()
    got = e.rf_predict_heap_quad(quads, roots, f.leaf_values, depth, X, 1)
    assert torch.allclose(got, ref, atol=1e-5), (got - ref).abs().max()


def test_conv_fp8_implicit_vs_im2col(ops):
    e = ops.ext()
    g = torch.Generator().manual_seed(7)
    N, H, W, C = 2, 14, 14, 32
    KO, R, stride, pad = 64, 3, 2, 1
    x = torch.randn(N, H, W, C, generator=g).to(DEV)
    w = torch.randn(KO, R * R * C, generator=g).to(DEV)
    xq = e.quantize_fp8(x, 1.0)
    wq = e.quantize_fp8(w, 1.0)
    got = e.conv_fp8(xq, wq, H, W, C, R, R, stride, pad, 1.0)
    aq = e.im2col_fp8(x, R, R, stride, pad, 1.0)
    # im2col quantizes from f32; quantize-then-gather must agree exactly here
    ref = e.gemm_fp8(aq, wq, 1.0)
    assert torch.allclose(got, ref, atol=1e-4), (got - ref).abs().max()


def test_conv_fp8_bias_relu_stride1(ops):
    e = ops.ext()
    g = torch.Generator().manual_seed(8)
    N, H, W, C = 1, 8, 8, 16
    KO, R = 32, 3
    x = torch.randn(N, H, W, C, generator=g).to(DEV)
    w = torch.randn(KO, R * R * C, generator=g).to(DEV)
    bias = torch.randn(KO, generator=g).to(DEV)
    xq, wq = e.quantize_fp8(x, 1.0), e.quantize_fp8(w, 1.0)
    got = e.conv_fp8(xq, wq, H, W, C, R, R, 1, 1, 1.0, bias, True)
    aq = e.im2col_fp8(x, R, R, 1, 1, 1.0)
    ref = e.gemm_fp8(aq, wq, 1.0, bias, True)
    assert torch.allclose(got, ref, atol=1e-4)


# ---- torch-free runtime (lpx_rt + standalone .hsaco) ---------------------

def test_lpx_rt_rf_matches_extension(ops):
    """hipModule path (no-torch runtime) must agree with the torch extension."""
    from lambdipy_amd.shims import lpx_rt

    n_trees, depth, F, C = 32, 8, 16, 4
    nodes_np, leaves_np = lpx_rt.random_heap_forest_np(n_trees, depth, F, C, 5)
    nodes = torch.from_numpy(nodes_np).to(DEV)
    leaves = torch.from_numpy(leaves_np).to(DEV)
    X = _rand((20000, F), seed=12)
    ref = ops.ext().rf_predict_heap(nodes, leaves, depth, X, 1)

    mod = lpx_rt.Module()
    out = torch.empty(20000, C, device=DEV)
    lpx_rt.rf_heap(mod, nodes.data_ptr(), leaves.data_ptr(), n_trees, depth,
                   X.data_ptr(), out.data_ptr(), 20000, F, C)
    lpx_rt.sync()
    assert torch.equal(out, ref)

    labels = torch.empty(20000, dtype=torch.int32, device=DEV)
    lpx_rt.argmax_rows(mod, out.data_ptr(), labels.data_ptr(), 20000, C)
    lpx_rt.sync()
    assert torch.equal(labels, ref.argmax(dim=1).to(torch.int32))


def test_lpx_rt_xgen_distribution(ops):
    from lambdipy_amd.shims import lpx_rt

    mod = lpx_rt.Module()
    n = 1 << 20
    x = torch.empty(n, device=DEV)
    lpx_rt.xgen(mod, x.data_ptr(), n, 7)
    lpx_rt.sync()
    assert torch.isfinite(x).all()
    assert abs(x.mean().item()) < 0.01
    assert abs(x.std().item() - 1.0) < 0.02


def test_lpx_rt_forestgen(ops):
    from lambdipy_amd.shims import lpx_rt

    mod = lpx_rt.Module()
    n_trees, depth, F, C = 8, 6, 12, 4
    n_int = n_trees * (2**depth - 1)
    n_rows = n_trees * 2**depth
    nodes = torch.empty(n_int, 2, dtype=torch.int32, device=DEV)
    leaves = torch.empty(n_rows, C, device=DEV)
    lpx_rt.forestgen(mod, nodes.data_ptr(), n_int, F, leaves.data_ptr(),
                     n_rows, C, 17)
    lpx_rt.sync()
    feats = nodes[:, 0]
    assert (feats >= 0).all() and (feats < F).all()
    thr = nodes[:, 1].view(torch.float32)
    assert torch.isfinite(thr).all()
    assert abs(thr.mean().item()) < 0.1
    s = leaves.sum(dim=1)
    # leaves are fp16-representable (device generator quantizes): rows sum
    # to 1 within fp16 rounding
    assert torch.allclose(s, torch.ones_like(s), atol=1e-3)
    # and thresholds round-trip fp16 exactly (compact-kernel contract)
    assert torch.equal(thr.to(torch.float16).to(torch.float32), thr)
    # the generated forest must traverse cleanly through the rf kernel
    X = _rand((4096, F), seed=3)
    out = torch.empty(4096, C, device=DEV)
    lpx_rt.rf_heap(mod, nodes.data_ptr(), leaves.data_ptr(), n_trees, depth,
                   X.data_ptr(), out.data_ptr(), 4096, F, C)
    lpx_rt.sync()
    ps = out.sum(dim=1) / n_trees
    assert torch.allclose(ps, torch.ones_like(ps), atol=1e-3)

    # checksum kernel vs torch sum
    labels = torch.randint(0, C, (100000,), dtype=torch.int32, device=DEV)
    ck = torch.zeros(1, dtype=torch.int64, device=DEV)
    lpx_rt.checksum_i32(mod, labels.data_ptr(), ck.data_ptr(), labels.numel())
    lpx_rt.sync()
    assert ck.item() == labels.sum().item()


# ---- shim-level parity (the layer-facing API surfaces) -------------------

def test_lpx_numpy_shim_matmul(ops):
    from lambdipy_amd.shims import lpx_numpy

    rng = np.random.default_rng(0)
    a = rng.standard_normal((256, 300)).astype(np.float32)
    b = rng.standard_normal((300, 192)).astype(np.float32)
    got = lpx_numpy.matmul(a, b)
    ref = a @ b
    rel = np.abs(got - ref).max() / (np.abs(ref).max() + 1e-6)
    assert rel < 2e-2, rel  # bf16 operand rounding


def test_lpx_numpy_shim_matvec(ops):
    from lambdipy_amd.shims import lpx_numpy

    rng = np.random.default_rng(1)
    a = rng.standard_normal((500, 700)).astype(np.float32)
    x = rng.standard_normal(700).astype(np.float32)
    got = lpx_numpy.matvec(a, x)
    ref = a @ x
    assert np.abs(got - ref).max() / (np.abs(ref).max() + 1e-6) < 2e-2


def test_lpx_scipy_shim_fft(ops):
    from lambdipy_amd.shims import lpx_scipy

    rng = np.random.default_rng(2)
    x = (rng.standard_normal((8, 512)) + 1j * rng.standard_normal((8, 512)))
    got = lpx_scipy.fft(x)
    ref = np.fft.fft(x.astype(np.complex64))
    assert np.abs(got - ref).max() / np.abs(ref).max() < 1e-4


def test_lpx_scipy_shim_csr(ops):
    scipy = pytest.importorskip("scipy")
    import scipy.sparse as sp
    from lambdipy_amd.shims import lpx_scipy

    rng = np.random.default_rng(3)
    A = sp.random(800, 600, density=0.02, random_state=4, format="csr",
                  dtype=np.float32)
    x = rng.standard_normal(600).astype(np.float32)
    got = lpx_scipy.csr_matvec(A.indptr, A.indices, A.data, x)
    ref = A @ x
    np.testing.assert_allclose(got, ref, rtol=1e-4, atol=1e-4)


def test_lpx_sklearn_shim_pairwise(ops):
    sklearn = pytest.importorskip("sklearn")
    from sklearn.metrics.pairwise import euclidean_distances
    from lambdipy_amd.shims import lpx_sklearn

    rng = np.random.default_rng(4)
    X = rng.standard_normal((300, 40)).astype(np.float32)
    Y = rng.standard_normal((200, 40)).astype(np.float32)
    got = lpx_sklearn.pairwise_sqeuclidean(X, Y)
    ref = euclidean_distances(X, Y, squared=True)
    assert np.abs(got - ref).max() < 1e-2


# ---- RF v8 compact pair (u16 feat + fp16 threshold, 8-B records) ----------

def _pack_pairs(n, depth):
    stride = (1 << depth) - 1
    T = n.shape[0] // stride
    nh = n.view(T, stride, 2)
    ppt = (1 << (depth - 1)) - 1
    idx = torch.arange(ppt, device=n.device)
    pairs = torch.cat([nh[:, 2 * idx + 1, :], nh[:, 2 * idx + 2, :]],
                      dim=2).reshape(T * ppt, 4).contiguous()
    roots = nh[:, 0, :].contiguous()
    thr16 = (nh[..., 1].contiguous().view(torch.float32).to(torch.float16)
             .view(torch.int16).to(torch.int32) & 0xFFFF)
    cn = (nh[..., 0] << 16) | thr16
    cpairs = (torch.stack([cn[:, 2 * idx + 1], cn[:, 2 * idx + 2]], dim=2)
              .reshape(T * ppt, 2).contiguous())
    return pairs, roots, cpairs, cn[:, 0].contiguous()


def test_rf_cpair_exact_on_fp16_forest(ops):
    """On a forest whose thresholds are fp16-representable, the compact
    kernel must be bitwise-identical to the f32 pair kernel."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    e = ops.ext()
    depth = 8
    f = ForestSoA.random(64, depth, 16, 8, seed=5, device=DEV)
    n = f.nodes.view(-1, 2).clone()
    n[:, 1] = (n[:, 1].view(torch.float32).to(torch.float16)
               .to(torch.float32).view(torch.int32))
    n = n.view_as(f.nodes).contiguous()
    X = _rand((8192, 16), seed=6)
    pairs, roots, cpairs, croots = _pack_pairs(n, depth)
    a = e.rf_predict_heap_pair(pairs, roots, f.leaf_values, depth, X, 1)
    b = e.rf_predict_heap_cpair(cpairs, croots, f.leaf_values, depth, X, 1)
    assert torch.equal(a, b), "compact kernel diverges on fp16 thresholds"
    b2 = e.rf_predict_heap_cpair(cpairs, croots, f.leaf_values, depth, X, 1)
    assert torch.equal(b, b2), "nondeterministic cpair kernel"


def test_rf_cpair_flip_rate_on_f32_forest(ops):
    """On an f32-threshold forest the fp16 rounding flips only decisions for
    x strictly between the two thresholds — the label-flip rate must be
    small (quantified error class)."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    e = ops.ext()
    depth = 10
    f = ForestSoA.random(128, depth, 32, 8, seed=7, device=DEV)
    # overwrite the (fp16-representable) synthetic thresholds with full-f32
    # randoms — the sklearn-fitted case this error class is about
    n = f.nodes.view(-1, 2)
    g = torch.Generator(device="cpu").manual_seed(77)
    thr = torch.randn(n.shape[0], generator=g).to(DEV)
    n[:, 1] = thr.view(torch.int32)
    X = _rand((65536, 32), seed=8)
    pairs, roots, cpairs, croots = _pack_pairs(f.nodes, depth)
    a = e.rf_predict_heap_pair(pairs, roots, f.leaf_values, depth, X, 1)
    b = e.rf_predict_heap_cpair(cpairs, croots, f.leaf_values, depth, X, 1)
    flips = (a.argmax(1) != b.argmax(1)).float().mean().item()
    # ~1e-3..1e-2 expected for N(0,1) thresholds at depth 10 x 128 trees
    assert flips < 0.05, f"label flip rate {flips} too high"
    # per-class sums stay close (each flipped path changes one leaf row)
    rel = (a - b).abs().max().item() / (a.abs().max().item() + 1e-6)
    assert rel < 0.2, f"proba divergence {rel}"


def test_rf_compact_api_routing(ops, monkeypatch):
    """Auto mode detects fp16-representable thresholds and routes through
    the compact kernel BITWISE-EXACTLY vs the f32 pair kernel; sklearn-like
    full-f32 thresholds fail the detection and keep the pair kernel."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    f = ForestSoA.random(32, 6, 12, 4, seed=9, device=DEV)
    X = _rand((4096, 12), seed=10)
    p1 = f.predict_proba_t(X)  # auto: synthetic forests are fp16-exact
    assert hasattr(f.nodes, "_lpx_cpair_cache"), "auto-compact not taken"
    monkeypatch.setenv("LPX_RF_COMPACT", "0")
    p2 = f.predict_proba_t(X)  # forced f32 pair path
    assert torch.equal(p1, p2), "compact path diverged on fp16-exact forest"
    monkeypatch.delenv("LPX_RF_COMPACT")
    # full-f32 thresholds: auto must NOT take the compact path
    n = f.nodes.view(-1, 2)
    g = torch.Generator(device="cpu").manual_seed(42)
    n[:, 1] = torch.randn(n.shape[0], generator=g).to(DEV).view(torch.int32)
    f.predict_proba_t(X)
    from lambdipy_amd.ops.api import _thresholds_fp16_exact
    assert not _thresholds_fp16_exact(f.nodes)


def test_spmv_xtile_vs_seg8(ops):
    """x-tile-in-LDS variant vs adaptive kernel on sorted-column CSR,
    including a shape with a non-multiple-of-tile column count."""
    for rows, cols, nnz, seed in ((100_000, 100_000, 32, 1),
                                  (5000, 40_000, 9, 2),
                                  (262_144, 50_000, 16, 3)):
        rng = np.random.default_rng(seed)
        idx = np.sort(rng.integers(0, cols, (rows, nnz)).astype(np.int32),
                      axis=1).reshape(-1)
        indptr = np.arange(0, (rows + 1) * nnz, nnz, dtype=np.int32)
        data = rng.standard_normal(rows * nnz).astype(np.float32)
        x = rng.standard_normal(cols).astype(np.float32)
        ti = torch.from_numpy(indptr).to(DEV)
        tj = torch.from_numpy(idx).to(DEV)
        td = torch.from_numpy(data).to(DEV)
        tx = torch.from_numpy(x).to(DEV)
        a = ops.spmv_csr(ti, tj, td, tx, -1)
        b = ops.spmv_csr(ti, tj, td, tx, 2)
        assert torch.allclose(a, b, rtol=1e-4, atol=1e-3), \
            f"xtile mismatch rows={rows}"


def test_rf_cpair_lv16_exact(ops):
    """fp16 leaf table: bitwise-identical to the f32-leaf compact kernel
    when leaf values are fp16-representable (the synthetic generators')."""
    from lambdipy_amd.shims.lpx_sklearn import ForestSoA

    e = ops.ext()
    depth = 8
    f = ForestSoA.random(64, depth, 16, 8, seed=21, device=DEV)
    X = _rand((16384, 16), seed=22)
    _, _, cpairs, croots = _pack_pairs(f.nodes, depth)
    a = e.rf_predict_heap_cpair(cpairs, croots, f.leaf_values, depth, X, 1)
    lv16 = f.leaf_values.to(torch.float16).contiguous()
    assert bool(lv16.to(torch.float32).eq(f.leaf_values).all().item()), \
        "generator must produce fp16-representable leaves"
    b = e.rf_predict_heap_cpair16(cpairs, croots, lv16, depth, X, 1)
    assert torch.equal(a, b)


# ---- non-power-of-two FFT (Bluestein over the pow-2 kernel) ---------------

@pytest.mark.parametrize("batch,n", [(4, 100), (8, 720), (2, 1000), (16, 97),
                                     (3, 1536), (1, 3), (4, 6000)])
def test_fft_nonpow2_vs_torch(ops, batch, n):
    g = torch.Generator(device="cpu").manual_seed(n)
    x = (torch.randn(batch, n, generator=g) +
         1j * torch.randn(batch, n, generator=g)).to(torch.complex64).to(DEV)
    got = ops.fft(x)
    ref = torch.fft.fft(x.to(torch.complex128))
    err = (got.to(torch.complex128) - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err / scale < 5e-4, f"n={n}: rel {err/scale}"
    # inverse round trip
    rt = ops.ifft(got)
    assert (rt - x).abs().max().item() / (x.abs().max().item() + 1e-6) < 1e-3


def test_lpx_rt_cpair16_matches_heap(ops):
    """standalone compact traversal (pair records + fp16 leaves + TILP4)
    must be BITWISE-equal to the f32 heap kernel on the same device-
    generated forest (shared per-index hash; leaves are fp16-exact)."""
    from lambdipy_amd.shims import lpx_rt

    mod = lpx_rt.Module()
    n_trees, depth, F, C = 33, 9, 24, 8  # odd tree count -> TILP tail
    stride = (1 << depth) - 1
    ppt = (1 << (depth - 1)) - 1
    n_int = n_trees * stride
    n_rows = n_trees * (1 << depth)
    nodes = torch.empty(n_int, 2, dtype=torch.int32, device=DEV)
    leaves = torch.empty(n_rows, C, device=DEV)
    lpx_rt.forestgen(mod, nodes.data_ptr(), n_int, F, leaves.data_ptr(),
                     n_rows, C, 71)
    cpairs = torch.empty(n_trees * ppt, 2, dtype=torch.int32, device=DEV)
    croots = torch.empty(n_trees, dtype=torch.int32, device=DEV)
    lv16 = torch.empty(n_rows, C, dtype=torch.int16, device=DEV)
    lpx_rt.forestgen_compact(mod, cpairs.data_ptr(), croots.data_ptr(),
                             n_trees, depth, F, lv16.data_ptr(), n_rows, C,
                             71)
    lpx_rt.sync()
    # compact node values must round-trip the heap generator's values
    heap = nodes.view(n_trees, stride, 2)
    thr16 = (heap[..., 1].contiguous().view(torch.float32)
             .to(torch.float16).view(torch.int16).to(torch.int32) & 0xFFFF)
    cn = (heap[..., 0] << 16) | thr16
    assert torch.equal(croots, cn[:, 0].to(torch.int32))
    assert torch.equal(lv16.view(torch.float16).to(torch.float32), leaves)
    X = _rand((20000, F), seed=72)
    out_heap = torch.empty(20000, C, device=DEV)
    out_c = torch.empty(20000, C, device=DEV)
    lpx_rt.rf_heap(mod, nodes.data_ptr(), leaves.data_ptr(), n_trees, depth,
                   X.data_ptr(), out_heap.data_ptr(), 20000, F, C)
    lpx_rt.rf_cpair16(mod, cpairs.data_ptr(), croots.data_ptr(),
                      lv16.data_ptr(), n_trees, depth, X.data_ptr(),
                      out_c.data_ptr(), 20000, F, C)
    lpx_rt.sync()
    assert torch.equal(out_heap, out_c), "compact standalone path diverges"

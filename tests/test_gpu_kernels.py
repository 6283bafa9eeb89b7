"""GPU kernel numerics tests: each HIP kernel vs a plain PyTorch/numpy fp32
reference of the same op, plus end-to-end GPU-vs-CPU engine equivalence."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs an MI355X", allow_module_level=True)


@pytest.fixture(scope="module")
def ext():
    from distributedkernelshap_amd.ops import load_extension

    return load_extension()


def test_extension_in_tree(ext):
    # the loaded .so must be the in-tree build (driver checks loaded paths)
    assert "distributedkernelshap_amd/ops" in ext.__file__


# ----------------------------------------------------------------------- #
# K2 sampler

def test_fill_random_masks_properties(ext):
    from distributedkernelshap_amd.core.sampler import plan_coalitions

    plan = plan_coalitions(12)  # S=2072, ne=596
    b, s, m = 4, plan.nsamples, 12
    ne = plan.enum_masks.shape[0]
    masks = torch.zeros(b, s, m, dtype=torch.uint8, device="cuda")
    cdf = torch.tensor(
        np.cumsum(plan.random_size_probs).astype(np.float32), device="cuda"
    )
    szs = torch.tensor(plan.random_sizes.astype(np.int32), device="cuda")
    ids = torch.arange(b, dtype=torch.int32, device="cuda")
    ext.fill_random_masks(masks, ne, plan.n_random, cdf, szs, 5, 0, ids)
    mh = masks.cpu().numpy()
    rnd = mh[:, ne:]
    sizes = rnd.sum(axis=2)
    # all rows filled with sizes from the residual range (draw 4-6, compl 6-8)
    assert sizes.min() >= 4 and sizes.max() <= 8
    # complement pairing within each fixed per-wave chunk (a pair never
    # crosses a chunk boundary; the last row of a chunk may be an unpaired
    # truncated draw)
    from distributedkernelshap_amd.core.sampler import sampler_chunks

    chunk = (plan.n_random + sampler_chunks(plan.n_random) - 1) \
        // sampler_chunks(plan.n_random)
    for bi in range(b):
        for clo in range(0, plan.n_random, chunk):
            chi = min(clo + chunk, plan.n_random)
            i = clo
            while i < chi - 1:
                if sizes[bi, i] <= 5:
                    assert np.array_equal(rnd[bi, i + 1], 1 - rnd[bi, i])
                    i += 2
                else:
                    i += 1
    # determinism + per-instance keying
    masks2 = torch.zeros_like(masks)
    ext.fill_random_masks(masks2, ne, plan.n_random, cdf, szs, 5, 0, ids)
    assert torch.equal(masks, masks2)
    assert not np.array_equal(mh[0], mh[1])


def test_fill_random_masks_sharding_invariance(ext):
    """Row b of a batch keyed [0..B) equals row 0 of a batch keyed [b] —
    the property the distributed shards rely on."""
    from distributedkernelshap_amd.core.sampler import plan_coalitions

    plan = plan_coalitions(12)
    ne = plan.enum_masks.shape[0]
    cdf = torch.tensor(
        np.cumsum(plan.random_size_probs).astype(np.float32), device="cuda"
    )
    szs = torch.tensor(plan.random_sizes.astype(np.int32), device="cuda")
    full = torch.zeros(4, plan.nsamples, 12, dtype=torch.uint8, device="cuda")
    ext.fill_random_masks(
        full, ne, plan.n_random, cdf, szs, 5, 0,
        torch.arange(4, dtype=torch.int32, device="cuda"),
    )
    single = torch.zeros(1, plan.nsamples, 12, dtype=torch.uint8, device="cuda")
    ext.fill_random_masks(
        single, ne, plan.n_random, cdf, szs, 5, 0,
        torch.tensor([2], dtype=torch.int32, device="cuda"),
    )
    assert torch.equal(full[2], single[0])


def test_fill_random_masks_size_distribution(ext):
    """The GPU sampler's subset-size draws follow the plan's residual
    Shapley-kernel distribution (frequency test against the supplied CDF —
    the CPU sampler has exact tests, this closes the loop for the device
    Philox stream)."""
    from distributedkernelshap_amd.core.sampler import plan_coalitions

    m = 20
    plan = plan_coalitions(m, 2 ** 14)
    assert plan.n_random > 4000
    b = 64
    ne = plan.enum_masks.shape[0]
    masks = torch.zeros(b, plan.nsamples, m, dtype=torch.uint8, device="cuda")
    cdf = torch.tensor(
        np.cumsum(plan.random_size_probs).astype(np.float32), device="cuda"
    )
    szs = torch.tensor(plan.random_sizes.astype(np.int32), device="cuda")
    ids = torch.arange(b, dtype=torch.int32, device="cuda")
    num_paired = int(np.floor((m - 1) / 2))
    ext.fill_random_masks(
        masks, ne, plan.n_random, cdf, szs, num_paired, 0, ids
    )
    sizes = masks[:, ne:].sum(dim=2).cpu().numpy().ravel()
    n = sizes.size
    # each draw of size s (prob p_s) emits one row of size s plus a
    # complement row of size m-s when s <= num_paired
    probs = {int(s): float(p) for s, p in
             zip(plan.random_sizes, plan.random_size_probs)}
    exp: dict = {}
    for s, p in probs.items():
        exp[s] = exp.get(s, 0.0) + p
        if s <= num_paired:
            exp[m - s] = exp.get(m - s, 0.0) + p
    zsum = sum(exp.values())
    for s in exp:
        exp[s] /= zsum
    # every emitted size must be an expected one
    assert set(np.unique(sizes)).issubset(set(exp))
    from distributedkernelshap_amd.core.sampler import sampler_chunks

    # ≤1 truncated complement per fixed per-wave chunk
    slack = sampler_chunks(plan.n_random) * b
    for s, p in exp.items():
        cnt = int((sizes == s).sum())
        sigma = (n * p * (1 - p)) ** 0.5
        assert abs(cnt - n * p) < 6 * sigma + slack, (
            s, cnt, n * p, sigma
        )


# ----------------------------------------------------------------------- #
# K3-K6 fused predict

def _fused_reference(masks, diff, base, wbg, act):
    """Plain torch fp32 reference of the fused kernel."""
    mf = masks.float()                                  # (B,S,M)
    b, s, m = mf.shape
    n_out, mpad, npad = diff.shape[1:]
    logits = torch.einsum("bsk,bokn->bson", mf, diff[:, :, :m]) + base[None, None]
    if act == 1:
        p = torch.sigmoid(logits)
    elif act == 2:
        p = torch.softmax(logits, dim=2)                # over o (dim 2 of bson)
    else:
        p = logits
    return torch.einsum("bson,n->bso", p, wbg)


@pytest.mark.parametrize("act", [0, 1, 2])
@pytest.mark.parametrize("n_out", [1, 2])
def test_fused_predict_linear_vs_torch(ext, act, n_out):
    if n_out == 1 and act == 2:
        pytest.skip("softmax needs n_out >= 2")
    g = torch.Generator(device="cuda").manual_seed(42)
    b, s, m, mpad, npad = 3, 200, 12, 12, 112
    n = 100
    masks = (torch.rand(b, s, m, generator=g, device="cuda") > 0.5).to(torch.uint8)
    diff = torch.zeros(b, n_out, mpad, npad, device="cuda")
    diff[:, :, :m, :n] = torch.randn(b, n_out, m, n, generator=g, device="cuda")
    base = torch.zeros(n_out, npad, device="cuda")
    base[:, :n] = torch.randn(n_out, n, generator=g, device="cuda")
    wbg = torch.zeros(npad, device="cuda")
    wbg[:n] = 1.0 / n
    ey = torch.empty(b, s, n_out, device="cuda")
    ext.fused_predict_linear(masks, diff, base, wbg, ey, act)
    ref = _fused_reference(masks, diff, base, wbg, act)
    assert torch.allclose(ey, ref, atol=2e-5, rtol=1e-4), (
        (ey - ref).abs().max().item()
    )


def test_fused_predict_nonmultiple_shapes(ext):
    """S not a multiple of 64, M not a multiple of 4, N not multiple of 16."""
    g = torch.Generator(device="cuda").manual_seed(7)
    b, s, m, n, n_out = 2, 130, 10, 37, 2
    mpad, npad = 12, 48
    masks = (torch.rand(b, s, m, generator=g, device="cuda") > 0.3).to(torch.uint8)
    diff = torch.zeros(b, n_out, mpad, npad, device="cuda")
    diff[:, :, :m, :n] = torch.randn(b, n_out, m, n, generator=g, device="cuda")
    base = torch.zeros(n_out, npad, device="cuda")
    base[:, :n] = torch.randn(n_out, n, generator=g, device="cuda")
    wbg = torch.zeros(npad, device="cuda")
    wbg[:n] = torch.rand(n, generator=g, device="cuda") + 0.1
    ey = torch.empty(b, s, n_out, device="cuda")
    ext.fused_predict_linear(masks, diff, base, wbg, ey, 2)
    ref = _fused_reference(masks, diff, base, wbg, 2)
    assert torch.allclose(ey, ref, atol=2e-5, rtol=1e-4)


# ----------------------------------------------------------------------- #
# K3' synth

def test_synth_chunk_vs_numpy(ext):
    g = torch.Generator(device="cuda").manual_seed(3)
    b, s, mg, n, d = 2, 40, 5, 16, 23
    masks = (torch.rand(b, s, mg, generator=g, device="cuda") > 0.5).to(torch.uint8)
    x = torch.randn(b, d, generator=g, device="cuda")
    bg = torch.randn(n, d, generator=g, device="cuda")
    colg = torch.randint(0, mg, (d,), generator=g, device="cuda", dtype=torch.int32)
    s_lo, s_hi = 10, 30
    out = torch.empty((s_hi - s_lo) * n, d, device="cuda")
    ext.synth_chunk(masks, x, bg, colg, out, 1, 2, s_lo, s_hi)
    mh, xh, bgh, cg = (t.cpu().numpy() for t in (masks, x, bg, colg))
    expect = np.empty(((s_hi - s_lo) * n, d), dtype=np.float32)
    for si in range(s_lo, s_hi):
        for ni in range(n):
            row = (si - s_lo) * n + ni
            take_x = mh[1, si, cg].astype(bool)
            expect[row] = np.where(take_x, xh[1], bgh[ni])
    assert np.array_equal(out.cpu().numpy(), expect)


# ----------------------------------------------------------------------- #
# K7 WLS

def test_wls_solve_vs_cpu(ext):
    from distributedkernelshap_amd.core.solver import solve_wls

    g = torch.Generator(device="cuda").manual_seed(11)
    b, s, m, n_out = 5, 500, 12, 2
    masks = (torch.rand(b, s, m, generator=g, device="cuda") > 0.5).to(torch.uint8)
    # avoid degenerate all-equal columns
    masks[:, :, 0] = (torch.rand(b, s, generator=g, device="cuda") > 0.3).to(torch.uint8)
    kw = torch.rand(b, s, generator=g, device="cuda") + 0.05
    ey = torch.randn(b, s, n_out, generator=g, device="cuda")
    total = torch.randn(b, n_out, generator=g, device="cuda")
    phi = torch.empty(b, m, n_out, device="cuda")
    # generic scalar kernel (no packed masks)
    ext.wls_solve(masks, kw, ey, total, phi)
    # MFMA Gram-build kernel (packed masks; (M-1)+n_out <= 16)
    packed = torch.empty(b, s, dtype=torch.int64, device="cuda")
    ext.pack_masks(masks, packed)
    phi_mfma = torch.empty_like(phi)
    ext.wls_solve(masks, kw, ey, total, phi_mfma, packed)
    ph = phi.cpu().numpy()
    ph_m = phi_mfma.cpu().numpy()
    for i in range(b):
        ref = solve_wls(
            masks[i].cpu().numpy(),
            kw[i].double().cpu().numpy(),
            ey[i].double().cpu().numpy(),
            total[i].double().cpu().numpy(),
        )
        assert np.allclose(ph[i], ref, atol=5e-3, rtol=1e-3), np.abs(ph[i] - ref).max()
        assert np.allclose(ph_m[i], ref, atol=5e-3, rtol=1e-3), np.abs(ph_m[i] - ref).max()
        # constraint holds exactly by construction
        assert np.allclose(ph[i].sum(axis=0), total[i].cpu().numpy(), atol=1e-4)
        assert np.allclose(ph_m[i].sum(axis=0), total[i].cpu().numpy(), atol=1e-4)


def test_wls_solve_m63(ext):
    from distributedkernelshap_amd.core.solver import solve_wls

    g = torch.Generator(device="cuda").manual_seed(13)
    b, s, m, n_out = 2, 3000, 63, 1
    masks = (torch.rand(b, s, m, generator=g, device="cuda") > 0.5).to(torch.uint8)
    kw = torch.rand(b, s, generator=g, device="cuda") + 0.05
    ey = torch.randn(b, s, n_out, generator=g, device="cuda")
    total = torch.randn(b, n_out, generator=g, device="cuda")
    phi = torch.empty(b, m, n_out, device="cuda")
    ext.wls_solve(masks, kw, ey, total, phi)
    ph = phi.cpu().numpy()
    for i in range(b):
        ref = solve_wls(
            masks[i].cpu().numpy(),
            kw[i].double().cpu().numpy(),
            ey[i].double().cpu().numpy(),
            total[i].double().cpu().numpy(),
        )
        assert np.allclose(ph[i], ref, atol=2e-2, rtol=5e-3), np.abs(ph[i] - ref).max()


# ----------------------------------------------------------------------- #
# end-to-end engine

def test_engine_gpu_vs_cpu_full_enumeration():
    """M=10 -> nsamples capped at 2^10-2 (full enumeration, zero sampling
    noise): GPU phi must match the fp64 CPU oracle to fp32 tolerance."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.Generator(np.random.Philox(key=[5, 5]))
    d = 10
    pred = LinearPredictor.random(d, 2, seed=5)
    bg = rng.normal(size=(50, d))
    X = rng.normal(size=(8, d))
    cpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    gpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    sv_c = cpu.shap_values(X)
    sv_g = gpu.shap_values(X)
    for o in range(2):
        assert np.allclose(sv_g[o], sv_c[o], atol=2e-4, rtol=1e-3), np.abs(
            sv_g[o] - sv_c[o]
        ).max()


def test_engine_gpu_local_accuracy_adult():
    """Adult-shaped config on GPU: sum(phi) == link(f(x)) - link(fnull)."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=32, n_background=100, seed=0)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=0)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = eng.shap_values(data.X)
    fx = logit(pred(data.X))
    for o in range(2):
        total = sv[o].sum(axis=1) + eng.expected_value[o]
        assert np.abs(total - fx[:, o]).max() < 1e-3


def test_engine_gpu_sharding_invariance():
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=8, n_background=50, seed=0)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=0)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    full = eng.shap_values(data.X)
    p1 = eng.shap_values(data.X[:4], instance_offset=0)
    p2 = eng.shap_values(data.X[4:], instance_offset=4)
    for o in range(2):
        assert np.allclose(
            full[o], np.concatenate([p1[o], p2[o]]), atol=1e-5
        )


def test_engine_gpu_torch_module_path():
    """Arbitrary-predictor path (synth_chunk + torch MLP) vs CPU oracle on a
    fully-enumerated config."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import TorchPredictor

    torch.manual_seed(0)
    d, n_out = 8, 2
    module = torch.nn.Sequential(
        torch.nn.Linear(d, 32), torch.nn.Tanh(), torch.nn.Linear(32, n_out),
        torch.nn.Softmax(dim=-1),
    ).double()
    module = module.float()
    pred = TorchPredictor(module, device="cuda")
    rng = np.random.Generator(np.random.Philox(key=[9, 9]))
    bg = rng.normal(size=(30, d))
    X = rng.normal(size=(4, d))
    cpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    gpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    sv_c = cpu.shap_values(X)
    sv_g = gpu.shap_values(X)
    for o in range(n_out):
        assert np.allclose(sv_g[o], sv_c[o], atol=5e-4, rtol=5e-3), np.abs(
            sv_g[o] - sv_c[o]
        ).max()


# ----------------------------------------------------------------------- #
# stress-config paths (library-GEMM fallback + batched torch WLS + module)

def test_engine_gpu_stress_paths():
    """M=200 > 64 dispatches to the tiled stress kernels
    (fused_predict_tiled + wls_gram); local accuracy and CPU-oracle
    agreement on a reduced stress shape."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import LinearPredictor, make_tabular

    data = make_tabular(n_features=200, n_instances=4, n_background=150, seed=1)
    pred = LinearPredictor.random(200, 2, seed=1)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = eng.shap_values(data.X, nsamples=2048, l1_reg=False)
    fx = logit(pred(data.X))
    for o in range(2):
        total = sv[o].sum(axis=1) + eng.expected_value[o]
        assert np.abs(total - fx[:, o]).max() < 2e-3
    # fp64 device-mode agreement: SAME device masks (counter RNG), double
    # arithmetic — isolates pipeline precision from sampling noise (a CPU
    # oracle would draw different random coalitions)
    from distributedkernelshap_amd.config import KernelConfig

    eng64 = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda", kernels=KernelConfig(predict_dtype="fp64"),
    )
    sv64 = eng64.shap_values(data.X, nsamples=2048, l1_reg=False)
    for o in range(2):
        err = np.abs(sv[o] - sv64[o]).max()
        assert err < 2e-3, err


def test_wls_gram_vs_fp64_reference(ext):
    """Tiled MFMA Gram+rhs build (stress WLS): matches the fp64 torch normal
    equations to chunked-promotion accuracy, across the M=257 word boundary
    and with kernel weights spanning orders of magnitude."""
    g = torch.Generator(device="cuda").manual_seed(17)
    for (b, s, m, n_out) in [(3, 2048, 100, 2), (2, 4096, 257, 2),
                             (2, 1000, 70, 1)]:
        masks = (torch.rand(b, s, m, generator=g, device="cuda") > 0.5).to(
            torch.uint8
        )
        # Shapley-like dynamic range
        kw = 10.0 ** (
            -4.0 * torch.rand(b, s, generator=g, device="cuda")
        )
        ey = torch.randn(b, s, n_out, generator=g, device="cuda")
        total = torch.randn(b, n_out, generator=g, device="cuda")
        w_words = (m + 63) // 64
        packed = torch.empty(b, s, w_words, dtype=torch.int64, device="cuda")
        ext.pack_masks_words(masks, packed)
        mm = m - 1
        a64 = torch.empty(b, mm, mm, dtype=torch.float64, device="cuda")
        r64 = torch.empty(b, mm, n_out, dtype=torch.float64, device="cuda")
        ext.wls_gram(packed, kw, ey, total, a64, r64)
        # fp64 reference
        z = masks.double()
        last = z[:, :, -1:]
        etmp = z[:, :, :-1] - last
        ey2 = ey.double() - last * total.double()[:, None, :]
        wz = etmp * kw.double()[:, :, None]
        a_ref = torch.bmm(wz.transpose(1, 2), etmp)
        r_ref = torch.bmm(wz.transpose(1, 2), ey2)
        ea = (a64 - a_ref).abs().max().item()
        er = (r64 - r_ref).abs().max().item()
        scale = a_ref.abs().max().item() + 1.0
        assert ea < 1e-4 * scale, (m, ea, scale)
        assert er < 1e-4 * scale, (m, er)


def test_fused_predict_tiled_vs_torch():
    """Tiled fused predict (M>64 / N>128) matches the library-GEMM fallback
    on the same device masks, for sigmoid, binary softmax and identity."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_tabular

    for act, n_out, m, n_bg in [("softmax", 2, 70, 200), ("sigmoid", 2, 150, 300),
                                ("none", 1, 80, 130)]:
        data = make_tabular(n_features=m, n_instances=3, n_background=n_bg,
                            seed=3)
        pred = LinearPredictor.random(m, n_out, seed=3, activation=act)
        eng = KernelShapEngine(
            pred, data.background, groups=data.groups, link="identity",
            seed=0, device="cuda",
        )
        gpu = eng._gpu
        varying = np.arange(m)
        plan = eng._plan(m, 1024)
        masks, _kw = gpu._device_masks(plan, np.arange(3))
        X_dev = torch.tensor(data.X, dtype=torch.float32, device="cuda")
        ey_tiled = gpu._ey_fused_tiled(masks, X_dev, varying).clone()
        ey_ref = gpu._ey_linear_torch(masks, X_dev, varying)
        err = (ey_tiled - ey_ref).abs().max().item()
        assert err < 5e-5, (act, n_out, m, n_bg, err)


def test_engine_gpu_mlp_module_path_local_accuracy():
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import make_predictor, make_tabular

    data = make_tabular(n_features=64, n_instances=4, n_background=50, seed=2)
    pred = make_predictor("mlp", 64, 2, seed=0, hidden=64, layers=2, device="cuda")
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = eng.shap_values(data.X, nsamples=1024, l1_reg=False)
    fx = logit(pred(data.X))
    for o in range(2):
        total = sv[o].sum(axis=1) + eng.expected_value[o]
        assert np.abs(total - fx[:, o]).max() < 2e-3


def test_engine_gpu_resnet_superpixels():
    """ResNet-18 superpixel config (BASELINE config 5), reduced: 112x112,
    nsamples=64."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import TorchPredictor
    from distributedkernelshap_amd.models.resnet import (
        make_superpixel_problem,
        resnet18,
    )

    X, bg, groups, names = make_superpixel_problem(n_instances=2, hw=112, patch=28)
    pred = TorchPredictor(resnet18(num_classes=4, seed=0), device="cuda")
    eng = KernelShapEngine(
        pred, bg, groups=groups, link="logit", seed=0, device="cuda"
    )
    sv = eng.shap_values(X, nsamples=64, l1_reg=False)
    assert sv[0].shape == (2, 16)
    fx = logit(pred(X))
    total = sv[0].sum(axis=1) + eng.expected_value[0]
    assert np.abs(total - fx[:, 0]).max() < 5e-3


def test_fill_random_masks_wide_m(ext):
    """M=200 (multi-word bitset path): sizes, pairing, determinism."""
    from distributedkernelshap_amd.core.sampler import plan_coalitions

    m = 200
    plan = plan_coalitions(m, nsamples=2048)
    ne = plan.enum_masks.shape[0]
    num_paired = int(np.floor((m - 1) / 2))
    cdf = torch.tensor(
        np.cumsum(plan.random_size_probs).astype(np.float32), device="cuda"
    )
    szs = torch.tensor(plan.random_sizes.astype(np.int32), device="cuda")
    masks = torch.zeros(2, plan.nsamples, m, dtype=torch.uint8, device="cuda")
    ids = torch.arange(2, dtype=torch.int32, device="cuda")
    ext.fill_random_masks(masks, ne, plan.n_random, cdf, szs, num_paired, 0, ids)
    mh = masks.cpu().numpy()
    rnd = mh[:, ne:]
    sizes = rnd.sum(axis=2)
    lo, hi = plan.random_sizes.min(), plan.random_sizes.max()
    assert sizes.min() >= lo and sizes.max() <= m - lo
    # paired draws followed by exact complements (within per-wave chunks)
    from distributedkernelshap_amd.core.sampler import sampler_chunks

    chunk = (plan.n_random + sampler_chunks(plan.n_random) - 1) \
        // sampler_chunks(plan.n_random)
    for bi in range(2):
        for clo in range(0, plan.n_random, chunk):
            chi = min(clo + chunk, plan.n_random)
            i = clo
            while i < chi - 1:
                if sizes[bi, i] <= num_paired:
                    assert np.array_equal(rnd[bi, i + 1], 1 - rnd[bi, i])
                    i += 2
                else:
                    i += 1
    masks2 = torch.zeros_like(masks)
    ext.fill_random_masks(masks2, ne, plan.n_random, cdf, szs, num_paired, 0, ids)
    assert torch.equal(masks, masks2)


def test_engine_tracing():
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=4, n_background=20, seed=0)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=0)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", device="cuda"
    )
    eng.enable_tracing()
    eng.shap_values(data.X)
    trace = eng.get_trace()
    for stage in ("varying", "bucket"):
        assert stage in trace and len(trace[stage]) >= 1
    # either the hipGraph fast path or the eager stage set must be present
    assert "graph" in trace or all(
        k in trace for k in ("masks", "predict", "wls", "d2h")
    )
    eng.enable_tracing(False)
    assert eng.get_trace() is None


@pytest.mark.skipif(
    __import__("os").environ.get("KSHAP_GRAPH") == "0",
    reason="graph capture disabled via env",
)
def test_graph_replay_tracks_new_inputs():
    """hipGraph replay must produce correct phi for NEW instances (the graph
    reads a static input buffer refreshed before each replay)."""
    import os

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=64, n_background=100, seed=4)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=4)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    X1, X2 = data.X[:32], data.X[32:]
    # call 1: eager (hit count 1); call 2: capture; call 3: replay with new X
    r1 = eng.shap_values(X1)
    r1b = eng.shap_values(X1)
    r1c = eng.shap_values(X1)
    assert np.allclose(r1[0], r1b[0], atol=1e-6) and np.allclose(
        r1[0], r1c[0], atol=1e-6
    )
    r2 = eng.shap_values(X2)  # replay with different inputs
    assert eng._gpu._graphs, "graph should have been captured"
    fx = logit(pred(X2))
    total = r2[0].sum(axis=1) + eng.expected_value[0]
    assert np.abs(total - fx[:, 0]).max() < 1e-3
    # replay result must differ from X1's (sanity that inputs propagated)
    assert not np.allclose(r2[0], r1[0])


def test_engine_gpu_mixed_varying_buckets():
    """Instances with different varying-group patterns exercise the
    multi-bucket GPU path (m==0, m==1 and two m>=2 buckets) and must match
    the CPU oracle."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.Generator(np.random.Philox(key=[21, 3]))
    d = 9
    pred = LinearPredictor.random(d, 2, seed=3)
    bg = rng.normal(size=(30, d))
    mu = bg.mean(axis=0)
    X = rng.normal(size=(6, d))
    X[1] = bg[0]              # m == 0 for the exact-background row? no: differs
    X[1, :] = bg[0, :]        # matches ONE row but differs from others -> varies
    X[2, 3:] = bg[:, 3:].mean(axis=0)  # still varies (mean != each row)
    # construct true non-varying columns: constant background columns
    bg[:, 0] = 1.5
    bg[:, 1] = -2.0
    X[3, 0] = 1.5             # group 0 fixed for instance 3
    X[4, 0] = 1.5
    X[4, 1] = -2.0            # groups 0,1 fixed for instance 4
    cpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    gpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    sv_c = cpu.shap_values(X)
    sv_g = gpu.shap_values(X)
    for o in range(2):
        assert np.allclose(sv_g[o], sv_c[o], atol=5e-4, rtol=1e-3), np.abs(
            sv_g[o] - sv_c[o]
        ).max()
    # instance 4's fixed groups get exactly zero attribution
    assert sv_g[0][4, 0] == 0.0 and sv_g[0][4, 1] == 0.0


def test_engine_gpu_l1_path():
    """Explicit l1_reg routes through the batched device LARS on GPU."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=4, n_background=30, seed=6)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=6)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = eng.shap_values(data.X, l1_reg="num_features(5)")
    assert sv[0].shape == (4, 12)
    assert (np.abs(sv[0]) > 1e-12).sum(axis=1).max() <= 6


def test_engine_gpu_l1_batched_matches_host_sklearn():
    """The batched device LARS (l1_device=True) selects the same features —
    and therefore the same phi — as the per-instance host sklearn path, for
    aic, num_features and float-alpha modes."""
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=5, n_background=30, seed=8)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=8)
    dev_eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    host_eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda", kernels=KernelConfig(l1_device=False),
    )
    for reg in ["aic", "num_features(4)", 0.01]:
        sv_d = dev_eng.shap_values(data.X, l1_reg=reg)
        sv_h = host_eng.shap_values(data.X, l1_reg=reg)
        for o in range(2):
            nz_d = np.abs(sv_d[o]) > 1e-12
            nz_h = np.abs(sv_h[o]) > 1e-12
            assert np.array_equal(nz_d, nz_h), (reg, o)
            assert np.allclose(sv_d[o], sv_h[o], atol=5e-3), (
                reg, o, np.abs(sv_d[o] - sv_h[o]).max()
            )


def test_gpu_pipeline_bitwise_deterministic():
    """Two identical runs produce bitwise-identical shap values — the
    share-nothing/stream-ordered design has no racy accumulation (SURVEY.md
    §5.2 rebuild requirement)."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=64, n_background=100, seed=8)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=8)

    def run():
        eng = KernelShapEngine(
            pred, data.background, groups=data.groups, link="logit", seed=0,
            device="cuda",
        )
        return eng.shap_values(data.X)

    a, b = run(), run()
    for o in range(2):
        assert np.array_equal(a[o], b[o])


# ----------------------------------------------------------------------- #
# bf16 matrix-core predict (opt-in predict_dtype)

def _bf16_engine(dtype):
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like

    data = make_adult_like(n_instances=32, n_background=100, seed=9)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=9)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda", kernels=KernelConfig(predict_dtype=dtype),
    )
    return eng, data, pred


def test_bf16x2_split_matches_fp32():
    """hi+lo split bf16 path is fp32-grade (error ~2^-16 on the B operand)."""
    eng32, data, pred = _bf16_engine("fp32")
    engb2, _, _ = _bf16_engine("bf16x2")
    sv32 = eng32.shap_values(data.X)
    svb2 = engb2.shap_values(data.X)
    for o in range(2):
        err = np.abs(sv32[o] - svb2[o]).max()
        assert err < 2e-3, err


def test_bf16_single_local_accuracy():
    """Plain bf16 path: coarser ey (~0.4% rel) but local accuracy still holds
    exactly (constrained solve) and phi stays close to fp32."""
    from distributedkernelshap_amd.core.links import logit

    engb, data, pred = _bf16_engine("bf16")
    sv = engb.shap_values(data.X)
    fx = logit(pred(data.X))
    for o in range(2):
        total = sv[o].sum(axis=1) + engb.expected_value[o]
        assert np.abs(total - fx[:, o]).max() < 1e-3
    eng32, _, _ = _bf16_engine("fp32")
    sv32 = eng32.shap_values(data.X)
    assert np.abs(sv[0] - sv32[0]).max() < 0.1


def test_build_diff_and_pack_kernels(ext):
    """Operand-layout builders vs plain torch references."""
    g = torch.Generator(device="cuda").manual_seed(17)
    b, G, O, n, m = 3, 9, 2, 20, 5
    mpad, npad = 8, 32
    xp = torch.randn(b, G, O, generator=g, device="cuda")
    bgp = torch.randn(n, G, O, generator=g, device="cuda")
    vidx = torch.tensor([0, 2, 3, 7, 8], dtype=torch.int64, device="cuda")
    out = torch.zeros(b, O, mpad, npad, device="cuda")
    ext.build_diff_f32(xp, bgp, vidx, out)
    ref = (xp[:, vidx].permute(0, 2, 1)[:, :, :, None]
           - bgp[:, vidx].permute(2, 1, 0)[None])          # (b, O, m, n)
    assert torch.allclose(out[:, :, :m, :n], ref, atol=1e-6)
    assert torch.all(out[:, :, m:, :] == 0) and torch.all(out[:, :, :, n:] == 0)

    outb = torch.zeros(b, 2, O, npad, 40, dtype=torch.bfloat16, device="cuda")
    ext.build_diff_bf16(xp, bgp, vidx, outb)
    recon = outb[:, 0].float() + outb[:, 1].float()        # hi + lo
    # (b, O, n, k) vs ref (b, O, k, n)
    assert torch.allclose(
        recon[:, :, :n, :m], ref.permute(0, 1, 3, 2), atol=1e-5
    )

    masks = (torch.rand(2, 30, 12, generator=g, device="cuda") > 0.5).to(torch.uint8)
    packed = torch.empty(2, 30, dtype=torch.int64, device="cuda")
    ext.pack_masks(masks, packed)
    mh, ph = masks.cpu().numpy(), packed.cpu().numpy().astype(np.uint64)
    for bi in range(2):
        for si in range(30):
            bits = sum(int(mh[bi, si, k]) << k for k in range(12))
            assert ph[bi, si] == bits


@pytest.mark.skipif(
    __import__("os").environ.get("KSHAP_GRAPH") == "0",
    reason="graph capture disabled via env",
)
def test_speculative_replay_fallback_on_pattern_change():
    """After the graph path engages (same shape twice), a batch whose varying
    pattern differs must be detected by the in-sync probe and answered by the
    eager path — with exact zeros for the fixed groups."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.Generator(np.random.Philox(key=[33, 1]))
    d = 8
    pred = LinearPredictor.random(d, 2, seed=1)
    bg = rng.normal(size=(40, d))
    bg[:, 0] = 0.5                      # constant background column
    X1 = rng.normal(size=(16, d))       # all groups vary
    eng = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    r1 = eng.shap_values(X1)            # eager (hit 1)
    r2 = eng.shap_values(X1)            # capture
    r3 = eng.shap_values(X1)            # speculative replay
    assert np.allclose(r1[0], r3[0], atol=1e-6)
    assert eng._gpu._spec is not None
    # same shape, different pattern: group 0 fixed for every instance
    X2 = rng.normal(size=(16, d))
    X2[:, 0] = 0.5
    r4 = eng.shap_values(X2)
    assert np.all(r4[0][:, 0] == 0.0) and np.all(r4[1][:, 0] == 0.0)
    # and local accuracy still holds on the fallback result
    from distributedkernelshap_amd.core.links import logit

    total = r4[0].sum(axis=1) + eng.expected_value[0]
    assert np.abs(total - logit(pred(X2))[:, 0]).max() < 1e-3
    # returning to the original pattern re-engages the graph path
    r5 = eng.shap_values(X1)
    assert np.allclose(r5[0], r1[0], atol=1e-6)


def test_engine_gpu_sigmoid_single_output():
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.Generator(np.random.Philox(key=[41, 2]))
    d = 7
    w = rng.normal(size=(1, d))
    pred = LinearPredictor(w, np.zeros(1), activation="sigmoid")
    bg = rng.normal(size=(1, d))   # single row: logit(sigmoid(z)) == z
    X = rng.normal(size=(8, d))
    eng = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    sv = eng.shap_values(X)
    assert len(sv) == 1
    expect = (X - bg[0]) * w[0]
    assert np.allclose(sv[0], expect, atol=5e-4), np.abs(sv[0] - expect).max()


def test_engine_gpu_fp64_mode_matches_oracle():
    """predict_dtype='fp64' runs the same device masks end-to-end in double:
    on a fully-enumerated plan it must match the CPU fp64 oracle to ~1e-12
    (same arithmetic, different order), making it a valid reference for the
    bench self-check."""
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.Generator(np.random.Philox(key=[9, 9]))
    d = 10
    pred = LinearPredictor.random(d, 2, seed=5)
    bg = rng.normal(size=(50, d))
    X = rng.normal(size=(6, d))
    cpu = KernelShapEngine(pred, bg, link="logit", seed=0, device="cpu")
    g64 = KernelShapEngine(
        pred, bg, link="logit", seed=0, device="cuda",
        kernels=KernelConfig(predict_dtype="fp64"),
    )
    sv_c = cpu.shap_values(X)
    sv_g = g64.shap_values(X)
    for o in range(2):
        err = np.abs(sv_g[o] - sv_c[o]).max()
        # the only fp32 artifacts left are the device X / background casts
        assert err < 1e-5, err
    # and the fp32 pipeline's error against it is small but NONZERO
    g32 = KernelShapEngine(pred, bg, link="logit", seed=0, device="cuda")
    sv_f = g32.shap_values(X)
    errs = [np.abs(sv_f[o] - sv_g[o]).max() for o in range(2)]
    assert max(errs) < 1e-3


def test_fused_predict_tiled_bf16_vs_fp32():
    """bf16x2 tiled predict (v_mfma_f32_16x16x32_bf16, hi+lo split) matches
    the f32 tiled kernel to fp32-grade tolerance on stress shapes."""
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_tabular

    for m, n_bg, act, n_out in [(200, 300, "softmax", 2), (96, 200, "sigmoid", 2)]:
        data = make_tabular(n_features=m, n_instances=3, n_background=n_bg,
                            seed=4)
        pred = LinearPredictor.random(m, n_out, seed=4, activation=act)
        eng = KernelShapEngine(
            pred, data.background, groups=data.groups, link="identity",
            seed=0, device="cuda",
            kernels=KernelConfig(predict_dtype="bf16x2"),
        )
        gpu = eng._gpu
        varying = np.arange(m)
        plan = eng._plan(m, 1024)
        masks, _kw = gpu._device_masks(plan, np.arange(3))
        X_dev = torch.tensor(data.X, dtype=torch.float32, device="cuda")
        ey_b = gpu._ey_fused_tiled_bf16(masks, X_dev, varying).clone()
        ey_f = gpu._ey_fused_tiled(masks, X_dev, varying).clone()
        err = (ey_b - ey_f).abs().max().item()
        assert err < 2e-3, (m, n_bg, act, err)


def test_engine_gpu_stress_bf16x2_local_accuracy():
    """End-to-end stress shape on the bf16x2 tiled path: local accuracy and
    fp32-grade agreement with the fp32 engine."""
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.core.links import logit
    from distributedkernelshap_amd.models import LinearPredictor, make_tabular

    data = make_tabular(n_features=200, n_instances=4, n_background=150, seed=1)
    pred = LinearPredictor.random(200, 2, seed=1)
    engb = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda", kernels=KernelConfig(predict_dtype="bf16x2"),
    )
    svb = engb.shap_values(data.X, nsamples=2048, l1_reg=False)
    fx = logit(pred(data.X))
    for o in range(2):
        total = svb[o].sum(axis=1) + engb.expected_value[o]
        assert np.abs(total - fx[:, o]).max() < 2e-3
    engf = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    svf = engf.shap_values(data.X, nsamples=2048, l1_reg=False)
    for o in range(2):
        err = np.abs(svb[o] - svf[o]).max()
        assert err < 5e-2, err

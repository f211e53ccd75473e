"""HIP kernel parity vs the fp32 CPU oracles (kernel tier).
All tests need an MI355X (marked gpu)."""
import numpy as np
import pytest
import torch

from g2vec_amd import ops
from g2vec_amd.ops import cpu_ref

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _csr(edges, weights, G):
    e = np.asarray(edges)
    order = np.lexsort((e[:, 1], e[:, 0]))
    e = e[order]
    w = np.asarray(weights, dtype=np.float32)[order]
    counts = np.bincount(e[:, 0], minlength=G)
    rp = np.zeros(G + 1, dtype=np.int32)
    rp[1:] = np.cumsum(counts)
    return (torch.from_numpy(rp), torch.from_numpy(e[:, 1].astype(np.int32)),
            torch.from_numpy(w))


def test_native_required_on_gpu(monkeypatch):
    """The GPU path must fail loudly without the extension, never fall back."""
    monkeypatch.setattr(ops, "_NATIVE", None)
    with pytest.raises(RuntimeError, match="HIP extension"):
        ops.native()


@pytest.mark.parametrize("gseed", [0, 5, 9])
def test_walks_bitwise_vs_cpu_oracle_uniform_weights(gseed):
    """With exactly-representable uniform weights the float sums are exact,
    so GPU and CPU walks follow identical RNG decisions bitwise."""
    rng = np.random.default_rng(gseed)
    G = 64
    edges = np.unique(rng.integers(0, G, size=(600, 2)), axis=0)
    edges = edges[edges[:, 0] != edges[:, 1]]
    rp, ci, w = _csr(edges, np.ones(len(edges)), G)
    src = torch.arange(G, dtype=torch.int32)
    cn, cl, ch = cpu_ref.random_walks(rp, ci, w, src, 4, 16, seed=123)
    gn, gl, gh = ops.random_walks(rp.to(DEV), ci.to(DEV), w.to(DEV),
                                  src.to(DEV), 4, 16, seed=123)
    assert torch.equal(gl.cpu(), cl)
    assert torch.equal(gn.cpu(), cn)
    assert torch.equal(gh.cpu(), ch)


def test_walks_structural_random_weights():
    rng = np.random.default_rng(1)
    G = 200
    edges = np.unique(rng.integers(0, G, size=(4000, 2)), axis=0)
    edges = edges[edges[:, 0] != edges[:, 1]]
    rp, ci, w = _csr(edges, rng.uniform(0.5, 1.0, len(edges)), G)
    src = torch.arange(G, dtype=torch.int32)
    nodes, lengths, hashes = ops.random_walks(rp.to(DEV), ci.to(DEV), w.to(DEV),
                                              src.to(DEV), 5, 40, seed=7)
    nodes, lengths = nodes.cpu().numpy(), lengths.cpu().numpy()
    adj = {(int(a), int(b)) for a, b in edges}
    for i in range(0, nodes.shape[0], 37):
        L = lengths[i]
        path = nodes[i, :L]
        assert nodes[i, 0] == i % G
        assert len(set(path.tolist())) == L
        for k in range(L - 1):
            assert (int(path[k]), int(path[k + 1])) in adj
    # hash parity with the CPU hash function
    for i in range(0, nodes.shape[0], 101):
        L = lengths[i]
        assert int(hashes[i]) == int(cpu_ref.path_hash(nodes[i, :L].tolist()))


def test_walks_sampling_distribution_gpu():
    edges = np.array([[0, 1], [0, 2], [0, 3]])
    rp, ci, w = _csr(edges, [0.6, 0.3, 0.1], 4)
    n = 20000
    nodes, _, _ = ops.random_walks(rp.to(DEV), ci.to(DEV), w.to(DEV),
                                   torch.zeros(1, dtype=torch.int32, device=DEV),
                                   n, 2, seed=3)
    first = nodes[:, 1].cpu().numpy()
    freq = np.bincount(first, minlength=4)[1:4] / n
    assert np.allclose(freq, [0.6, 0.3, 0.1], atol=0.02)


def _pathset_tensors(G=300, P=500, seed=2):
    rng = np.random.default_rng(seed)
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 30))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    return (torch.tensor(genes, dtype=torch.int32),
            torch.tensor(offs, dtype=torch.int32),
            torch.tensor(labels))


def test_cbow_fwd_scalar_matches_oracle():
    genes, offs, labels = _pathset_tensors()
    s = torch.randn(300)
    lc, cc, dc = cpu_ref.cbow_fwd_scalar(s, genes, offs, labels, 1 / 500, True)
    lg, cg, dg = ops.cbow_fwd_scalar(s.to(DEV), genes.to(DEV), offs.to(DEV),
                                     labels.to(DEV), 1 / 500, True)
    assert torch.allclose(lg.cpu(), lc, atol=1e-5)
    assert torch.equal(cg.cpu(), cc)
    assert torch.allclose(dg.cpu(), dc, atol=1e-7)


def test_scatter_do_det_matches_oracle():
    genes, offs, labels = _pathset_tensors(seed=3)
    dO = torch.randn(500)
    c_ref = cpu_ref.scatter_dO(genes, offs, dO, 300)
    plan = ops.build_scatter_plan(genes.to(DEV), offs.to(DEV), 300)
    c_gpu = ops.scatter_dO(genes.to(DEV), offs.to(DEV), dO.to(DEV), 300, plan)
    assert torch.allclose(c_gpu.cpu(), c_ref, atol=1e-5)
    # determinism: two runs bitwise equal
    c2 = ops.scatter_dO(genes.to(DEV), offs.to(DEV), dO.to(DEV), 300, plan)
    assert torch.equal(c_gpu, c2)


def test_scatter_slab_blocked_matches_unblocked():
    """Slab-blocked scatter (one launch per L2-sized dO slice, ascending
    accumulate into c) must match the single-pass reduce and be
    deterministic across runs."""
    genes, offs, labels = _pathset_tensors(seed=9)
    dO = torch.randn(500)
    g, o, d = genes.to(DEV), offs.to(DEV), dO.to(DEV)
    c_ref = ops.scatter_dO(g, o, d, 300)        # single-slab (500 paths)
    orig = ops.SLAB_BYTES
    try:
        ops.SLAB_BYTES = 4 * 64                 # force ~8 slabs of 64 paths
        plan = ops.build_scatter_plan(g, o, 300)
        assert plan.slab_seg_ptr is not None and len(plan.slab_seg_ptr) > 2
        c1 = ops.scatter_dO(g, o, d, 300, plan)
        c2 = ops.scatter_dO(g, o, d, 300, plan)
    finally:
        ops.SLAB_BYTES = orig
    assert torch.equal(c1, c2)                  # deterministic
    assert torch.allclose(c1, c_ref, atol=1e-5)
    assert torch.allclose(c1.cpu(), cpu_ref.scatter_dO(genes, offs, dO, 300),
                          atol=1e-5)


def test_adam_kernels_match_oracle():
    torch.manual_seed(0)
    G, h = 128, 128
    c = torch.randn(G)
    who = torch.randn(h)
    W = torch.randn(G, h)
    m = torch.rand(G, h) * 0.1
    v = torch.rand(G, h) * 0.1
    Wg, mg, vg = (x.clone().to(DEV) for x in (W, m, v))
    cpu_ref.adam_rank1(W, m, v, c, who, 5, 0.005, 0.9, 0.999, 1e-8)
    ops.adam_rank1(Wg, mg, vg, c.to(DEV), who.to(DEV), 5, 0.005, 0.9, 0.999, 1e-8)
    assert torch.allclose(Wg.cpu(), W, atol=1e-6)
    assert torch.allclose(mg.cpu(), m, atol=1e-7)
    assert torch.allclose(vg.cpu(), v, atol=1e-7)

    grad = torch.randn(G, h)
    W2, m2, v2 = torch.randn(G, h), torch.zeros(G, h), torch.zeros(G, h)
    W2g, m2g, v2g = (x.clone().to(DEV) for x in (W2, m2, v2))
    cpu_ref.adam_dense(W2, m2, v2, grad, 1, 0.005, 0.9, 0.999, 1e-8)
    ops.adam_dense(W2g, m2g, v2g, grad.to(DEV), 1, 0.005, 0.9, 0.999, 1e-8)
    assert torch.allclose(W2g.cpu(), W2, atol=1e-6)


@pytest.mark.parametrize("hidden", [64, 128, 256, 1024])
@pytest.mark.parametrize("dtype", ["fp32", "bf16", "fp16"])
def test_cbow_fwd_general_matches_oracle(hidden, dtype):
    genes, offs, labels = _pathset_tensors(seed=4)
    torch.manual_seed(1)
    W = torch.randn(300, hidden)
    who = torch.randn(hidden)
    Wd = W.to(DEV)
    if dtype == "bf16":
        Wd = Wd.bfloat16().contiguous()
    elif dtype == "fp16":
        Wd = Wd.half().contiguous()
    Wc = Wd.cpu().float()   # oracle sees the same (possibly rounded) weights
    lc, cc, dc, Hc = cpu_ref.cbow_fwd(Wc, who, genes, offs, labels, 1 / 500, True)
    lg, cg, dg, Hg = ops.cbow_fwd(Wd, who.to(DEV), genes.to(DEV), offs.to(DEV),
                                  labels.to(DEV), 1 / 500, True)
    atol = 1e-4 if dtype == "fp32" else 5e-3
    assert torch.allclose(Hg.cpu(), Hc, atol=atol)
    assert torch.allclose(lg.cpu(), lc, atol=atol)
    assert (cg.cpu() == cc).float().mean() > 0.99   # ties can flip at bf16
    assert torch.allclose(dg.cpu(), dc, atol=atol)


def test_cbow_bwd_rows_matches_oracle():
    genes, offs, labels = _pathset_tensors(seed=5)
    who = torch.randn(128)
    dO = torch.randn(500) * 0.01
    ref = cpu_ref.cbow_bwd_rows(who, genes, offs, dO, 300)
    got = ops.cbow_bwd_rows(who.to(DEV), genes.to(DEV), offs.to(DEV),
                            dO.to(DEV), 300)
    assert torch.allclose(got.cpu(), ref, atol=1e-5)


def test_pcc_edges_matches_corrcoef():
    rng = np.random.default_rng(6)
    G, S = 50, 77
    X = rng.standard_normal((S, G)).astype(np.float32)
    expr = torch.from_numpy(X)
    from g2vec_amd.graph import zscore_group
    zt = zscore_group(expr, torch.zeros(S, dtype=torch.int64), 0)
    edges = torch.tensor([[i, j] for i in range(10) for j in range(10) if i != j],
                         dtype=torch.int32)
    w = ops.pcc_edges(zt.to(DEV), edges.to(DEV), S).cpu().numpy()
    R = np.corrcoef(X.T)
    ref = np.array([abs(R[i, j]) for i, j in edges.numpy()])
    assert np.allclose(w, ref, atol=1e-4)


def test_corr_gemm_matches_torch_mm():
    rng = np.random.default_rng(7)
    for G, S in ((100, 77), (257, 135)):
        zt = torch.from_numpy(rng.standard_normal((G, S)).astype(np.float32))
        C_ref = (zt @ zt.t()) / S
        C = ops.corr_gemm(zt.to(DEV), S).cpu()
        assert torch.allclose(C, C_ref, atol=1e-4), (G, S)


def test_bf16_copy():
    x = torch.randn(1000, device=DEV)
    assert torch.equal(ops.native().bf16_copy(x), x.bfloat16())


def test_cbow_eval_counts_matches_oracle():
    genes, offs, labels = _pathset_tensors(seed=8)
    s = torch.randn(300)
    counts = torch.zeros(2, device=DEV)
    ops.cbow_eval_counts_(s.to(DEV), genes.to(DEV), offs.to(DEV),
                          labels.to(DEV), 300, counts)
    _l, corr, _d = cpu_ref.cbow_fwd_scalar(s, genes, offs, labels, 1.0, False)
    assert float(counts[0]) == pytest.approx(float(corr[:300].sum()))
    assert float(counts[1]) == pytest.approx(float(corr[300:].sum()))


def test_cbow_eval_counts_lds_variant_bitwise(monkeypatch):
    """The opt-in LDS-staged eval (cbow_eval_counts_lds_kernel,
    G2VEC_EVAL_LDS=<bytes>) must be bitwise-equal to the default subwave
    kernel: same gathers, same subwave-16 accumulation order, s staged in
    LDS. Off by default (measured in-chain neutral, profiles/README.md)."""
    genes, offs, labels = _pathset_tensors(seed=9)
    s = torch.randn(300)
    dO_a = torch.zeros(300, device=DEV)
    dO_b = torch.zeros(300, device=DEV)
    counts_a = torch.zeros(2, device=DEV)
    counts_b = torch.zeros(2, device=DEV)
    args = (s.to(DEV), genes.to(DEV), offs.to(DEV), labels.to(DEV), 300)
    monkeypatch.delenv("G2VEC_EVAL_LDS", raising=False)
    ops.cbow_eval_counts_(*args, counts_a, dO=dO_a, inv_b=1.0 / 300)
    monkeypatch.setenv("G2VEC_EVAL_LDS", str(64 * 1024))
    for grid in ("512", "1024"):
        monkeypatch.setenv("G2VEC_EVAL_LDS_GRID", grid)
        counts_b.zero_()
        dO_b.zero_()
        ops.cbow_eval_counts_(*args, counts_b, dO=dO_b, inv_b=1.0 / 300)
        assert torch.equal(counts_a, counts_b)
        assert torch.equal(dO_a, dO_b)


@pytest.mark.parametrize("G,h", [(1000, 64), (777, 128), (513, 256), (300, 512)])
def test_gemv_kernels_match_torch(G, h):
    torch.manual_seed(3)
    W = torch.randn(G, h, device=DEV)
    x = torch.randn(h, device=DEV)
    c = torch.randn(G, device=DEV)
    s = torch.empty(G, device=DEV)
    ops.gemv_rows(W, x, s)
    assert torch.allclose(s, torch.mv(W, x), atol=1e-4)
    g = torch.empty(h, device=DEV)
    ops.gemv_cols(W, c, g)
    assert torch.allclose(g, torch.mv(W.t(), c), atol=1e-3)


def test_walks_bitwise_high_degree_paths():
    """Exercises all three sampling paths (lean <=64, register multichunk
    <=256, chunked fallback >256) bitwise vs the CPU oracle: a dense graph
    with a >256-degree hub, uniform (exactly representable) weights."""
    rng = np.random.default_rng(42)
    G = 400
    edges = set()
    for d in range(G):              # hub 0: degree ~G-1 (> 256)
        if d != 0:
            edges.add((0, d))
    for s in range(1, G):           # mid-degree rows (~100)
        for d in rng.choice(G, size=100, replace=False):
            if int(d) != s:
                edges.add((s, int(d)))
    edges = np.array(sorted(edges))
    rp, ci, w = _csr(edges, np.ones(len(edges)), G)
    src = torch.arange(G, dtype=torch.int32)
    cn, cl, ch = cpu_ref.random_walks(rp, ci, w, src, 2, 10, seed=77)
    gn, gl, gh = ops.random_walks(rp.to(DEV), ci.to(DEV), w.to(DEV),
                                  src.to(DEV), 2, 10, seed=77)
    assert torch.equal(gl.cpu(), cl)
    assert torch.equal(gn.cpu(), cn)
    assert torch.equal(gh.cpu(), ch)


@pytest.mark.timeout(600)
def test_eval_scan_matches_subwave_kernel_and_oracle():
    """Instance-parallel segmented-scan eval (eval_scan/finish kernels)
    vs the subwave-per-path kernel and the CPU oracle: identical correct
    counts, dO within fp32 reduction-order tolerance. Covers paths that
    span multiple 64-instance windows and window-straddling boundaries."""
    import numpy as np

    from g2vec_amd.ops import cpu_ref
    rng = np.random.default_rng(31)
    G, P = 500, 3000
    genes, offs, labels = [], [0], []
    for p in range(P):
        # mix of tiny and window-spanning paths (up to 150 genes)
        L = int(rng.integers(1, 150)) if p % 7 == 0 else int(rng.integers(1, 25))
        L = min(L, G)
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda")
    s = torch.randn(G, device=dev) * 0.3
    g_t = torch.tensor(genes, dtype=torch.int32, device=dev)
    o_t = torch.tensor(offs, dtype=torch.int32, device=dev)
    l_t = torch.tensor(labels, dtype=torch.float32, device=dev)
    p_split = P * 4 // 5
    inv_b = 1.0 / p_split

    counts_a = torch.zeros(2, device=dev)
    dO_a = torch.zeros(p_split, device=dev)
    ops.native().cbow_eval_counts_(s, g_t, o_t, l_t, p_split, counts_a,
                                   dO=dO_a, inv_b=inv_b)

    lens = (o_t[1:] - o_t[:-1]).long()
    pathid = torch.repeat_interleave(
        torch.arange(P, dtype=torch.int32, device=dev), lens)
    cap = 150 // 64 + 2
    piece = torch.empty(P * cap, dtype=torch.float32, device=dev)
    counts_b = torch.zeros(2, device=dev)
    dO_b = torch.zeros(p_split, device=dev)
    ops.native().cbow_eval_scan_(s, g_t, pathid, o_t, l_t, p_split, cap,
                                 piece, counts_b, dO=dO_b, inv_b=inv_b)

    assert torch.equal(counts_a.cpu(), counts_b.cpu())
    assert torch.allclose(dO_a, dO_b, atol=1e-6)

    # oracle cross-check
    _l, corr, d = cpu_ref.cbow_fwd_scalar(s.cpu(), g_t.cpu(), o_t.cpu(),
                                          l_t.cpu(), inv_b, True)
    assert float(counts_b[0]) == float(corr[:p_split].sum())
    assert float(counts_b[1]) == float(corr[p_split:].sum())
    assert torch.allclose(dO_b.cpu(), d[:p_split], atol=1e-5)


@pytest.mark.timeout(600)
def test_relu_general_kernels_match_oracle_on_gpu():
    """--activation relu GPU kernels (fwd act + masked backward) vs the
    CPU oracle."""
    import numpy as np

    from g2vec_amd.ops import cpu_ref
    rng = np.random.default_rng(19)
    G, P, h = 300, 500, 128
    genes, offs, labels = [], [0], []
    for _ in range(P):
        L = int(rng.integers(1, 20))
        genes += rng.choice(G, size=L, replace=False).tolist()
        offs.append(offs[-1] + L)
        labels.append(float(rng.integers(0, 2)))
    dev = torch.device("cuda")
    W = torch.randn(G, h, device=dev) * 0.1
    who = torch.randn(h, device=dev) * 0.1
    g_t = torch.tensor(genes, dtype=torch.int32, device=dev)
    o_t = torch.tensor(offs, dtype=torch.int32, device=dev)
    l_t = torch.tensor(labels, dtype=torch.float32, device=dev)
    inv_b = 1.0 / P
    loss_g, corr_g, dO_g, H_g = ops.cbow_fwd(W, who, g_t, o_t, l_t, inv_b,
                                             True, act=1)
    dW_g = ops.cbow_bwd_rows(who, g_t, o_t, dO_g, G, H_pre=H_g)
    loss_c, corr_c, dO_c, H_c = cpu_ref.cbow_fwd(
        W.cpu(), who.cpu(), g_t.cpu(), o_t.cpu(), l_t.cpu(), inv_b, True,
        act=1)
    dW_c = cpu_ref.cbow_bwd_rows(who.cpu(), g_t.cpu(), o_t.cpu(), dO_c, G,
                                 H_pre=H_c)
    assert torch.allclose(loss_g.cpu(), loss_c, atol=1e-5)
    assert torch.equal(corr_g.cpu(), corr_c)
    assert torch.allclose(H_g.cpu(), H_c, atol=1e-5)
    assert torch.allclose(dW_g.cpu(), dW_c, atol=1e-5)


@pytest.mark.timeout(600)
def test_adam_rank1_fused_matches_unfused_sequence():
    """The fused epoch tail (rank-1 Adam + in-pass dW_ho partials + who
    fold/update) vs the explicit gemv_cols + adam_rank1 + adam_dense
    sequence: W/m/v bitwise (same math), who/mO/vO within fp32
    reduction-order tolerance (the dW_ho summation tree differs)."""
    dev = torch.device("cuda")
    for h in (64, 128, 512, 1024):
        G = 5000
        gen = torch.Generator(device="cpu").manual_seed(h)
        W0 = torch.randn(G, h, generator=gen).to(dev) * 0.1
        m0 = torch.randn(G, h, generator=gen).to(dev) * 0.01
        v0 = torch.rand(G, h, generator=gen).to(dev) * 0.01
        c = torch.randn(G, generator=gen).to(dev) * 0.1
        who0 = torch.randn(h, generator=gen).to(dev) * 0.1
        mO0 = torch.randn(h, generator=gen).to(dev) * 0.01
        vO0 = torch.rand(h, generator=gen).to(dev) * 0.01
        lrt = torch.tensor([0.003], device=dev)

        Wa, ma, va = W0.clone(), m0.clone(), v0.clone()
        whoa, mOa, vOa = who0.clone(), mO0.clone(), vO0.clone()
        gw = torch.empty(h, device=dev)
        ops.native().gemv_cols_(Wa, c, gw)
        ops.native().adam_rank1(Wa, ma, va, c, whoa, lrt, 0.9, 0.999, 1e-8)
        ops.native().adam_dense(whoa, mOa, vOa, gw, lrt, 0.9, 0.999, 1e-8)

        Wb, mb, vb = W0.clone(), m0.clone(), v0.clone()
        whob, mOb, vOb = who0.clone(), mO0.clone(), vO0.clone()
        ops.native().adam_rank1(Wb, mb, vb, c, whob, lrt, 0.9, 0.999, 1e-8,
                                mO=mOb, vO=vOb)
        assert torch.equal(Wa, Wb) and torch.equal(ma, mb)
        assert torch.equal(va, vb)
        assert torch.allclose(whoa, whob, atol=1e-6)
        assert torch.allclose(mOa, mOb, atol=1e-6)
        assert torch.allclose(vOa, vOb, atol=1e-6)
        # determinism: repeat the fused call from the same state
        Wc, mc, vc = W0.clone(), m0.clone(), v0.clone()
        whoc, mOc, vOc = who0.clone(), mO0.clone(), vO0.clone()
        ops.native().adam_rank1(Wc, mc, vc, c, whoc, lrt, 0.9, 0.999, 1e-8,
                                mO=mOc, vO=vOc)
        assert torch.equal(whob, whoc) and torch.equal(mOb, mOc)

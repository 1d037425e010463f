"""HIP kernel numerics vs plain-torch fp32 references (reference test
style: tests/cpp_extensions/*).  All tests require an MI355X."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs GPU", allow_module_level=True)

from realhf_amd.ops import functional as F  # noqa: E402
import realhf_amd._C as C  # noqa: E402 — must exist on a GPU box


def test_mfma_fragment_layout():
    """Asymmetric A/B probe: validates the assumed 16x16x32 lane mapping."""
    torch.manual_seed(0)
    A = torch.randn(16, 32) * 0.5
    B = torch.randn(32, 16) * 0.5
    D = C.mfma_probe(A.cuda(), B.cuda()).cpu()
    ref = A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float()
    torch.testing.assert_close(D, ref, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("rows,H", [(7, 128), (640, 4096), (1000, 11008 // 2 * 2)])
def test_rmsnorm_fwd_bwd(rows, H):
    torch.manual_seed(1)
    x = torch.randn(rows, H, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(H, dtype=torch.bfloat16, device="cuda")
    x_ref = x.clone().float().requires_grad_(True)
    w_ref = w.clone().float().requires_grad_(True)
    ref = F.rms_norm_ref(x_ref, w_ref, 1e-5)
    xg = x.clone().requires_grad_(True)
    wg = w.clone().requires_grad_(True)
    out = F._RMSNormFn.apply(xg, wg, 1e-5)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)
    g = torch.randn_like(ref)
    ref.backward(g)
    out.backward(g.to(torch.bfloat16))
    torch.testing.assert_close(xg.grad.float(), x_ref.grad, atol=1e-1, rtol=1e-1)
    torch.testing.assert_close(
        wg.grad.float(), w_ref.grad, atol=0.5, rtol=5e-2
    )


def test_rope():
    torch.manual_seed(2)
    total, nh, hd = 300, 8, 128
    x = torch.randn(total, nh, hd, dtype=torch.bfloat16, device="cuda")
    pos = torch.randint(0, 500, (total,), device="cuda")
    cos, sin = F.rotary_cache.get(hd, 512, 10000.0, torch.device("cuda"))
    out = C.rope_fwd(x, cos, sin, pos.long(), False, False)
    ref = F.apply_rotary_ref(x.float(), cos, sin, pos, False)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    # conj is the inverse rotation
    back = C.rope_fwd(out, cos, sin, pos.long(), False, True)
    torch.testing.assert_close(back.float(), x.float(), atol=5e-2, rtol=5e-2)


def test_swiglu_fwd_bwd():
    torch.manual_seed(3)
    gu = torch.randn(257, 2 * 2816, dtype=torch.bfloat16, device="cuda")
    ref_in = gu.clone().float().requires_grad_(True)
    ref = F.swiglu_ref(ref_in)
    out = C.swiglu_fwd(gu)
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    g = torch.randn_like(ref)
    ref.backward(g)
    dgu = C.swiglu_bwd(g.to(torch.bfloat16), gu)
    torch.testing.assert_close(dgu.float(), ref_in.grad, atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("bs", [1, 4, 16])
def test_gae(bs):
    rng = np.random.RandomState(bs)
    lens = rng.randint(3, 50, size=bs)
    total = int(lens.sum())
    cu = np.concatenate([[0], np.cumsum(lens)])
    rewards = torch.randn(total, device="cuda")
    values = torch.randn(total + bs, device="cuda")
    bootstrap = torch.tensor(rng.rand(bs) > 0.5, device="cuda")
    cut = torch.tensor(cu, dtype=torch.int32, device="cuda")
    adv, ret = C.gae_1d(rewards, values, cut, bootstrap, 0.99, 0.95)
    adv_ref, ret_ref = F.gae_ref(
        rewards.cpu(), values.cpu(), cut.cpu(), bootstrap.cpu(), 0.99, 0.95
    )
    torch.testing.assert_close(adv.cpu(), adv_ref, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(ret.cpu(), ret_ref, atol=1e-4, rtol=1e-4)


def test_intervals():
    torch.manual_seed(4)
    n = 10_000_000
    src = torch.randn(n, dtype=torch.bfloat16, device="cuda")
    rng = np.random.RandomState(0)
    starts = np.sort(rng.choice(n - 2000, size=500, replace=False))
    ivs = []
    for s in starts:
        ivs.append((int(s), int(s + rng.randint(1, 1500))))
    ivs = F.merge_intervals(ivs)
    iv_t = torch.tensor(ivs, dtype=torch.long, device="cuda")
    out = C.slice_intervals(src, iv_t)
    ref = F.slice_intervals_ref(src, iv_t.cpu())
    assert torch.equal(out.cpu(), ref.cpu())
    dst = torch.zeros_like(src)
    C.set_intervals(out, dst, iv_t)
    dst_ref = torch.zeros_like(src.cpu())
    F.set_intervals_ref(out.cpu(), dst_ref, iv_t.cpu())
    assert torch.equal(dst.cpu(), dst_ref)


def test_fused_adamw():
    torch.manual_seed(5)
    n = 4096 * 8
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda") * 0.1
    m = torch.randn(n, device="cuda").abs() * 0.01
    v = torch.randn(n, device="cuda").abs() * 0.001
    bf = torch.zeros(n, dtype=torch.bfloat16, device="cuda")
    p2, m2, v2 = p.clone(), m.clone(), v.clone()
    C.fused_adamw(p, g, m, v, bf, 1e-3, 0.9, 0.95, 1e-5, 0.1, 7, 1.0, True)
    F.fused_adamw_ref(p2, g, m2, v2, 1e-3, 0.9, 0.95, 1e-5, 0.1, 7)
    torch.testing.assert_close(p, p2, atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(m, m2, atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(v, v2, atol=1e-7, rtol=1e-5)
    torch.testing.assert_close(bf.float(), p2, atol=1e-2, rtol=1e-2)
    # bf16 grads + clip scale
    p3, m3, v3 = p.clone(), m.clone(), v.clone()
    p4, m4, v4 = p.clone(), m.clone(), v.clone()
    gb = g.to(torch.bfloat16)
    C.fused_adamw(p3, gb, m3, v3, bf, 1e-3, 0.9, 0.95, 1e-5, 0.1, 8, 0.5, True)
    F.fused_adamw_ref(p4, gb, m4, v4, 1e-3, 0.9, 0.95, 1e-5, 0.1, 8,
                      grad_scale=0.5)
    torch.testing.assert_close(p3, p4, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(m3, m4, atol=1e-6, rtol=1e-5)


@pytest.mark.parametrize("nq,nkv", [(8, 8), (8, 2)])
@pytest.mark.parametrize("lens", [[64], [128, 64, 200], [1, 17, 330]])
def test_attn_varlen_fwd(nq, nkv, lens):
    torch.manual_seed(6)
    hd = 128
    total = sum(lens)
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32, device="cuda")
    q = torch.randn(total, nq, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    k = torch.randn(total, nkv, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    v = torch.randn(total, nkv, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    scale = 1.0 / np.sqrt(hd)
    out, lse = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0)
    ref = F.attn_varlen_ref(
        q.float().cpu(), k.float().cpu(), v.float().cpu(), cu.cpu(), True, scale
    )
    torch.testing.assert_close(out.float().cpu(), ref, atol=3e-2, rtol=3e-2)
    assert torch.isfinite(lse[:, 0]).all()


@pytest.mark.parametrize("nq,nkv,hd", [(32, 32, 128), (8, 2, 128), (8, 8, 64)])
def test_attn_decode(nq, nkv, hd):
    torch.manual_seed(7)
    bs, maxlen = 5, 400
    lens = torch.tensor([1, 50, 200, 399, 64], dtype=torch.int32, device="cuda")
    q = torch.randn(bs, nq, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    kc = torch.randn(bs, maxlen, nkv, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    vc = torch.randn(bs, maxlen, nkv, hd, dtype=torch.bfloat16, device="cuda") * 0.5
    scale = 1.0 / np.sqrt(hd)
    out = C.attn_decode(q, kc, vc, lens, scale, 0)
    ref = F.attn_decode_ref(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), lens.cpu(), scale
    )
    torch.testing.assert_close(out.float().cpu(), ref, atol=3e-2, rtol=3e-2)


def test_attn_varlen_autograd_bwd():
    """The autograd path (HIP fwd + recompute bwd) vs full torch fp32."""
    torch.manual_seed(8)
    lens = [96, 130]
    total = sum(lens)
    nq = nkv = 4
    hd = 128
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32, device="cuda")
    q = (torch.randn(total, nq, hd, device="cuda") * 0.5).to(torch.bfloat16)
    k = (torch.randn(total, nkv, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v = (torch.randn(total, nkv, hd, device="cuda") * 0.5).to(torch.bfloat16)
    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    vg = v.clone().requires_grad_(True)
    out = F.attn_varlen(qg, kg, vg, cu, max(lens), True, None)
    g = torch.randn_like(out)
    out.backward(g)
    qr = q.float().cpu().requires_grad_(True)
    kr = k.float().cpu().requires_grad_(True)
    vr = v.float().cpu().requires_grad_(True)
    ref = F.attn_varlen_ref(qr, kr, vr, cu.cpu(), True, 1.0 / np.sqrt(hd))
    ref.backward(g.float().cpu())
    torch.testing.assert_close(out.float().cpu(), ref, atol=4e-2, rtol=4e-2)
    torch.testing.assert_close(qg.grad.float().cpu(), qr.grad, atol=8e-2, rtol=8e-2)
    torch.testing.assert_close(kg.grad.float().cpu(), kr.grad, atol=8e-2, rtol=8e-2)
    torch.testing.assert_close(vg.grad.float().cpu(), vr.grad, atol=8e-2, rtol=8e-2)


def test_fused_decode_layer_matches_eager():
    """The fused rope_qkv_decode path vs the composed eager decode path."""
    import os

    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models.real_model import ReaLModel

    fam = hf_reg.get_family("llama")
    cfg = fam.make_test_config(
        n_layers=2, hidden_dim=256, n_heads=4, n_kv_heads=2, head_dim=64,
        vocab_size=128,
    )
    torch.manual_seed(11)
    model = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    model.random_init()
    bs, maxlen = 3, 32
    kv = [
        (
            torch.zeros(bs, maxlen, 2, 64, dtype=torch.bfloat16, device="cuda"),
            torch.zeros(bs, maxlen, 2, 64, dtype=torch.bfloat16, device="cuda"),
        )
        for _ in range(cfg.n_layers)
    ]
    kv2 = [(k.clone(), v.clone()) for k, v in kv]
    tokens = torch.randint(0, 128, (bs,), device="cuda")
    cache_seqlens = torch.tensor([5, 9, 2], dtype=torch.int32, device="cuda")
    with torch.no_grad():
        out_fused = model(
            packed_input_ids=tokens, kv_caches=kv, cache_seqlens=cache_seqlens,
            decode=True,
        )
        os.environ["REALHF_AMD_NO_FUSED_DECODE"] = "1"
        try:
            out_eager = model(
                packed_input_ids=tokens, kv_caches=kv2,
                cache_seqlens=cache_seqlens, decode=True,
            )
        finally:
            del os.environ["REALHF_AMD_NO_FUSED_DECODE"]
    torch.testing.assert_close(out_fused.float(), out_eager.float(),
                               atol=3e-2, rtol=3e-2)
    for (k1, v1), (k2, v2) in zip(kv, kv2):
        torch.testing.assert_close(k1.float(), k2.float(), atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(v1.float(), v2.float(), atol=2e-2, rtol=2e-2)


def test_grouped_gemm():
    """MFMA grouped GEMM vs per-expert torch GEMMs."""
    torch.manual_seed(12)
    E, N, K = 4, 256, 192
    lens = torch.tensor([70, 0, 129, 33], dtype=torch.int32)
    total = int(lens.sum())
    x = (torch.randn(total, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(E, N, K, device="cuda") * 0.5).to(torch.bfloat16)
    out = C.grouped_gemm(x, w, lens)
    off = 0
    for e in range(E):
        n = int(lens[e])
        if n == 0:
            continue
        ref = (x[off:off + n].float() @ w[e].float().t())
        torch.testing.assert_close(
            out[off:off + n].float(), ref, atol=0.3, rtol=3e-2
        )
        off += n


def test_moe_grouped_path_matches_loop():
    """MoELayer grouped-GEMM inference path vs the per-expert loop."""
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models.real_model import ReaLModel

    fam = hf_reg.get_family("mixtral")
    cfg = fam.make_test_config(
        n_layers=1, hidden_dim=64, n_heads=1, n_kv_heads=1, vocab_size=128,
        intermediate_dim=128, head_dim=64,
    )
    torch.manual_seed(13)
    m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    m.random_init()
    m.eval()
    toks = torch.randint(0, 128, (96,), device="cuda")
    cu = torch.tensor([0, 48, 96], dtype=torch.int32, device="cuda")
    with torch.no_grad():
        out_grouped = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=48)
        os.environ["REALHF_AMD_MOE_LOOP"] = "1"  # force per-expert loop ref
        try:
            out_loop = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=48)
        finally:
            del os.environ["REALHF_AMD_MOE_LOOP"]
    torch.testing.assert_close(
        out_grouped.float(), out_loop.float(), atol=5e-2, rtol=5e-2
    )


def test_add_rmsnorm_fused():
    torch.manual_seed(14)
    x = torch.randn(40, 4096, dtype=torch.bfloat16, device="cuda")
    res = torch.randn(40, 4096, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
    out, s = C.add_rmsnorm_fwd(x, res, w, 1e-5)
    s_ref = (x.float() + res.float()).to(torch.bfloat16)
    ref = F.rms_norm_ref(s_ref.float(), w.float(), 1e-5)
    torch.testing.assert_close(s.float(), s_ref.float(), atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(out.float(), ref, atol=5e-2, rtol=5e-2)


def test_skinny_gemm_with_residual():
    torch.manual_seed(15)
    M, K, N = 16, 11008 // 32 * 32, 4096
    x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    res = (torch.randn(M, N, device="cuda") * 0.3).to(torch.bfloat16)
    ws = torch.empty(32 * M * N, dtype=torch.float32, device="cuda")
    ref = x.float() @ w.float().t()
    out = C.skinny_gemm(x, w, ws, 16, None)
    torch.testing.assert_close(out.float(), ref, atol=0.3, rtol=3e-2)
    out_r = C.skinny_gemm(x, w, ws, 16, res)
    torch.testing.assert_close(out_r.float(), ref + res.float(), atol=0.3,
                               rtol=3e-2)
    # M < 16 tail
    x2 = x[:5]
    out2 = C.skinny_gemm(x2.contiguous(), w, ws, 8, None)
    torch.testing.assert_close(out2.float(), ref[:5], atol=0.3, rtol=3e-2)


def test_skinny_gemm2_fused_combine():
    """v2 (semaphore-fused split-K combine) vs fp32 reference, repeated
    launches to exercise the self-resetting semaphores."""
    torch.manual_seed(16)
    sem = torch.zeros(1024, dtype=torch.int32, device="cuda")
    for (M, K, N, sk) in [(16, 4096, 12288, 8), (16, 4096, 4096, 8),
                          (16, 11008, 4096, 16), (5, 4096, 4096, 8)]:
        x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
        res = (torch.randn(M, N, device="cuda") * 0.3).to(torch.bfloat16)
        ws = torch.empty(32 * 16 * N, dtype=torch.float32, device="cuda")
        ref = x.float() @ w.float().t()
        for rep in range(3):  # reuse => semaphores must be back at zero
            out = C.skinny_gemm2(x, w, ws, sem, sk, None)
            torch.testing.assert_close(out.float(), ref, atol=0.3, rtol=3e-2)
        out_r = C.skinny_gemm2(x, w, ws, sem, sk, res)
        torch.testing.assert_close(out_r.float(), ref + res.float(),
                                   atol=0.3, rtol=3e-2)
        assert int(sem.abs().sum()) == 0  # all counters self-reset


def test_slab_consumers_match_combined():
    """skinny_gemm_nc slabs consumed in rope/add_rmsnorm/swiglu prologues
    (launch-boundary reduce) match the combine-then-consume path."""
    torch.manual_seed(17)
    M, H = 16, 4096
    ws = torch.empty(32 * 16 * 32000, dtype=torch.float32, device="cuda")

    # --- add_rmsnorm slab mode
    wo = (torch.randn(H, H, device="cuda") * 0.05).to(torch.bfloat16)
    attn = (torch.randn(M, H, device="cuda") * 0.3).to(torch.bfloat16)
    x = (torch.randn(M, H, device="cuda") * 0.3).to(torch.bfloat16)
    g = torch.randn(H, device="cuda").to(torch.bfloat16)
    o_bf = C.skinny_gemm(attn, wo, ws, 8, None)
    ref_h2, ref_x2 = C.add_rmsnorm_fwd(o_bf, x, g, 1e-5)
    parts = C.skinny_gemm_nc(attn, wo, ws, 8)
    assert parts.shape == (8, M, H)
    h2, x2 = C.add_rmsnorm_fwd(parts, x, g, 1e-5)
    torch.testing.assert_close(x2.float(), ref_x2.float(), atol=0.1, rtol=5e-2)
    torch.testing.assert_close(h2.float(), ref_h2.float(), atol=0.1, rtol=5e-2)

    # --- swiglu slab mode
    I = 11008
    wgu = (torch.randn(2 * I, H, device="cuda") * 0.05).to(torch.bfloat16)
    gu_bf = C.skinny_gemm(h2, wgu, ws, 8, None)
    ref_act = C.swiglu_fwd(gu_bf)
    gu_parts = C.skinny_gemm_nc(h2, wgu, ws, 8)
    act = C.swiglu_fwd(gu_parts)
    torch.testing.assert_close(act.float(), ref_act.float(), atol=0.1,
                               rtol=5e-2)

    # --- rope_qkv_decode slab mode (bs=16, llama-7B head geometry)
    nq = nkv = 8
    hd = 128
    qkvd = (nq + 2 * nkv) * hd
    wqkv = (torch.randn(qkvd, H, device="cuda") * 0.05).to(torch.bfloat16)
    hin = (torch.randn(M, H, device="cuda") * 0.3).to(torch.bfloat16)
    maxlen = 64
    kc1 = torch.zeros(M, maxlen, nkv, hd, dtype=torch.bfloat16, device="cuda")
    vc1 = torch.zeros_like(kc1)
    kc2 = torch.zeros_like(kc1)
    vc2 = torch.zeros_like(kc1)
    lens = torch.randint(1, maxlen, (M,), dtype=torch.int32, device="cuda")
    pos = torch.arange(maxlen, device="cuda", dtype=torch.float32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, hd, 2, device="cuda").float() / hd))
    ang = pos[:, None] * inv[None, :]
    cosb, sinb = ang.cos().contiguous(), ang.sin().contiguous()
    qkv_bf = C.skinny_gemm(hin, wqkv, ws, 8, None)
    q_ref = C.rope_qkv_decode(qkv_bf, None, kc1, vc1, lens, cosb, sinb, nq, True)
    qkv_parts = C.skinny_gemm_nc(hin, wqkv, ws, 8)
    q = C.rope_qkv_decode(qkv_parts, None, kc2, vc2, lens, cosb, sinb, nq, True)
    torch.testing.assert_close(q.float(), q_ref.float(), atol=0.1, rtol=5e-2)
    torch.testing.assert_close(kc2.float(), kc1.float(), atol=0.1, rtol=5e-2)
    torch.testing.assert_close(vc2.float(), vc1.float(), atol=0.1, rtol=5e-2)


def test_attn_varlen_bwd_hip_matches_ref():
    """Hand-written MFMA backward vs fp32 autograd reference (packed
    varlen causal, mixed lengths, MHA and GQA)."""
    from realhf_amd.ops import functional as F

    torch.manual_seed(21)
    for nq, nkv in ((8, 8), (8, 2)):
        hd = 128
        lens = [96, 64, 130, 32]
        total = sum(lens)
        cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                          dtype=torch.int32, device="cuda")
        q = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
        k = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        v = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        dout = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
        scale = hd ** -0.5

        # fp32 reference grads via autograd over the blocked torch impl
        qr = q.float().requires_grad_(True)
        kr = k.float().requires_grad_(True)
        vr = v.float().requires_grad_(True)
        ref = F._attn_varlen_blocked_torch(qr, kr, vr, cu, True, scale)
        ref_dq, ref_dk, ref_dv = torch.autograd.grad(
            ref, (qr, kr, vr), dout.float())

        # HIP fwd for lse, then HIP bwd
        out, lse = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0)
        dsum = (dout.float() * out.float()).sum(-1)
        dq32, dk32, dv32 = C.attn_varlen_bwd(q, k, v, dout.contiguous(),
                                             lse, dsum, cu, True, scale, 0)
        rep = nq // nkv
        if rep > 1:
            dk32 = dk32.view(total, nkv, rep, hd).sum(2)
            dv32 = dv32.view(total, nkv, rep, hd).sum(2)
        torch.testing.assert_close(dv32, ref_dv, atol=0.15, rtol=5e-2)
        torch.testing.assert_close(dk32, ref_dk, atol=0.15, rtol=5e-2)
        torch.testing.assert_close(dq32, ref_dq, atol=0.15, rtol=5e-2)


def test_attn_varlen_fwd_sliding_window():
    """MFMA forward with a binding sliding window vs the fp32 oracle."""
    from realhf_amd.ops import functional as F

    torch.manual_seed(23)
    nq = nkv = 8
    hd = 128
    for window in (32, 100):
        lens = [96, 64, 200, 40]
        cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                          dtype=torch.int32, device="cuda")
        total = sum(lens)
        q = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
        k = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        v = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        scale = hd ** -0.5
        out, _ = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, window)
        ref = F.attn_varlen_ref(q.float().cpu(), k.float().cpu(),
                                v.float().cpu(), cu.cpu(), True, scale,
                                window=window)
        torch.testing.assert_close(out.float().cpu(), ref, atol=0.05,
                                   rtol=3e-2)
        # and the window must differ from full causal
        out_full, _ = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0)
        assert not torch.allclose(out, out_full)


def test_attn_decode_sliding_window():
    from realhf_amd.ops import functional as F

    torch.manual_seed(25)
    bs, nq, nkv, hd, maxlen = 4, 8, 8, 128, 96
    q = (torch.randn(bs, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
    kc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
    vc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
    lens = torch.tensor([96, 40, 7, 64], dtype=torch.int32, device="cuda")
    scale = hd ** -0.5
    for window in (16, 48):
        out = C.attn_decode(q, kc, vc, lens, scale, window)
        ref = F.attn_decode_ref(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), lens.cpu(), scale,
                                window=window)
        torch.testing.assert_close(out.float().cpu(), ref, atol=0.05,
                                   rtol=3e-2)
    full = C.attn_decode(q, kc, vc, lens, scale, 0)
    assert not torch.allclose(C.attn_decode(q, kc, vc, lens, scale, 16), full)


def test_grouped_gemm_backward_kernels():
    """dX/dW MFMA kernels vs fp32 torch references (per expert)."""
    torch.manual_seed(14)
    E, N, K = 4, 256, 192
    lens = torch.tensor([70, 0, 129, 33], dtype=torch.int32)
    total = int(lens.sum())
    x = (torch.randn(total, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(E, N, K, device="cuda") * 0.5).to(torch.bfloat16)
    dout = (torch.randn(total, N, device="cuda") * 0.5).to(torch.bfloat16)
    dx = C.grouped_gemm_dx(dout, w, lens)
    dw = C.grouped_gemm_dw(dout, x, lens, E)
    off = 0
    for e in range(E):
        n = int(lens[e])
        dw_ref = (dout[off:off + n].float().t() @ x[off:off + n].float()
                  if n else torch.zeros(N, K, device="cuda"))
        torch.testing.assert_close(dw[e].float(), dw_ref, atol=0.5, rtol=3e-2)
        if n:
            dx_ref = dout[off:off + n].float() @ w[e].float()
            torch.testing.assert_close(dx[off:off + n].float(), dx_ref,
                                       atol=0.5, rtol=3e-2)
        off += n


def test_moe_training_grouped_backward_matches_loop():
    """A full MoE train forward+backward on the native grouped path must
    produce the same flat-buffer gradients as the per-expert loop path
    (VERDICT round-1 item #4: Mixtral GRPO trains off the native kernel
    without this)."""
    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models.real_model import ReaLModel

    fam = hf_reg.get_family("mixtral")
    cfg = fam.make_test_config(
        n_layers=1, hidden_dim=64, n_heads=1, n_kv_heads=1, vocab_size=128,
        intermediate_dim=128, head_dim=64,
    )

    def run(force_loop):
        torch.manual_seed(15)
        m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
        m.random_init()
        m.train()
        m.allocate_grad_buffer()
        for k, p in m._params.items():
            p.requires_grad_(True)
            p.grad = m.grad_view(k)  # autograd accumulates into flat_grad
        toks = torch.randint(0, 128, (96,), device="cuda")
        cu = torch.tensor([0, 48, 96], dtype=torch.int32, device="cuda")
        if force_loop:
            os.environ["REALHF_AMD_MOE_LOOP"] = "1"
        try:
            out = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=48)
            out.float().square().mean().backward()
        finally:
            os.environ.pop("REALHF_AMD_MOE_LOOP", None)
        return out.detach().float(), m.flat_grad.detach().float().clone()

    out_g, grad_g = run(force_loop=False)
    out_l, grad_l = run(force_loop=True)
    torch.testing.assert_close(out_g, out_l, atol=5e-2, rtol=5e-2)
    # bf16 kernels vs bf16 loop: small elementwise tolerance on grads
    torch.testing.assert_close(grad_g, grad_l, atol=8e-2, rtol=8e-2)


def test_attn_varlen_bwd_sliding_window():
    """MFMA backward with a binding sliding window vs the fp32 autograd
    oracle (closes the round-1 gap: windowed models previously trained
    through a torch recompute fallback)."""
    from realhf_amd.ops import functional as F

    torch.manual_seed(27)
    for nq, nkv, window in ((8, 8, 32), (8, 2, 100)):
        hd = 128
        lens = [96, 64, 200, 40]
        total = sum(lens)
        cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                          dtype=torch.int32, device="cuda")
        q = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
        k = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        v = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
        dout = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
        scale = hd ** -0.5

        qr = q.float().requires_grad_(True)
        kr = k.float().requires_grad_(True)
        vr = v.float().requires_grad_(True)
        ref = F._attn_varlen_blocked_torch(qr, kr, vr, cu, True, scale,
                                           window=window)
        ref_dq, ref_dk, ref_dv = torch.autograd.grad(
            ref, (qr, kr, vr), dout.float())

        out, lse = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale,
                                     window)
        dsum = (dout.float() * out.float()).sum(-1)
        dq32, dk32, dv32 = C.attn_varlen_bwd(q, k, v, dout.contiguous(),
                                             lse, dsum, cu, True, scale,
                                             window)
        rep = nq // nkv
        if rep > 1:
            dk32 = dk32.view(total, nkv, rep, hd).sum(2)
            dv32 = dv32.view(total, nkv, rep, hd).sum(2)
        torch.testing.assert_close(dv32, ref_dv, atol=0.15, rtol=5e-2)
        torch.testing.assert_close(dk32, ref_dk, atol=0.15, rtol=5e-2)
        torch.testing.assert_close(dq32, ref_dq, atol=0.15, rtol=5e-2)
        # the windowed grads must differ from full-causal grads
        dq_f, _, _ = C.attn_varlen_bwd(q, k, v, dout.contiguous(), lse,
                                       dsum, cu, True, scale, 0)
        assert not torch.allclose(dq32, dq_f)


def test_offloaded_adamw_matches_resident():
    """Pipelined host-offloaded AdamW (chunked H2D/compute/D2H on a side
    stream) must produce the same parameters as the GPU-resident
    optimizer (70B-tier machinery, VERDICT round-1 item #6)."""
    from realhf_amd.models.hf.llama import make_test_config
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel.ddp import OptimizerConfig, ZeRO1Optimizer

    os.environ["REALHF_AMD_OFFLOAD_CHUNK"] = "100000"  # force many chunks

    def run(offload):
        torch.manual_seed(33)
        cfg = make_test_config(n_layers=2, hidden_dim=128, n_heads=2,
                               n_kv_heads=2, head_dim=64,
                               intermediate_dim=256, vocab_size=256)
        cfg.family = "llama"
        m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
        m.random_init()
        opt = ZeRO1Optimizer(
            m, OptimizerConfig(lr=1e-2, warmup_steps_proportion=0.0,
                               offload=offload),
            total_train_steps=10,
        )
        for _ in range(3):  # multi-step: state round-trips must persist
            opt.zero_grad()
            toks = torch.randint(0, 256, (64,), device="cuda")
            cu = torch.tensor([0, 64], dtype=torch.int32, device="cuda")
            out = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=64)
            out.float().square().mean().backward()
            opt.step()
        torch.cuda.synchronize()
        return m.flat_param.detach().clone()

    try:
        p_resident = run(offload=False)
        p_offload = run(offload=True)
    finally:
        del os.environ["REALHF_AMD_OFFLOAD_CHUNK"]
    torch.testing.assert_close(p_offload, p_resident, atol=0, rtol=0)


def test_decode_graph_capture_failure_falls_back_eager():
    """hipGraph capture failure (e.g. a graph-unsafe op such as an
    in-graph collective) must fall back to eager decode with identical
    tokens (VERDICT round-1 item #2: explicit tested fallback)."""
    from realhf_amd.api.model import GenerationHyperparameters
    from realhf_amd.models.generation import generate
    from realhf_amd.models.hf.llama import make_test_config
    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(n_layers=2, hidden_dim=128, n_heads=2,
                           n_kv_heads=2, head_dim=64, intermediate_dim=256,
                           vocab_size=256)
    torch.manual_seed(44)
    m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    m.random_init()
    toks = torch.randint(0, 256, (24,), device="cuda")
    cu = torch.tensor([0, 12, 24], dtype=torch.int32, device="cuda")
    g = GenerationHyperparameters(max_new_tokens=8, greedy=True,
                                  use_hip_graph=True)
    out_graph = generate(m, toks, cu, g)
    m._gen_session = None  # fresh session for the failing-capture run
    os.environ["REALHF_AMD_FORCE_GRAPH_FAIL"] = "1"
    try:
        out_eager = generate(m, toks, cu, g)
    finally:
        del os.environ["REALHF_AMD_FORCE_GRAPH_FAIL"]
    assert torch.equal(out_graph.gen_tokens, out_eager.gen_tokens)

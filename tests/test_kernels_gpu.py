"""HIP kernel numerics tests (MI355X, @gpu): each csrc kernel vs a plain
PyTorch fp32 reference computed FROM THE SAME bf16-rounded inputs, so the
only difference is accumulation order — tolerances are tight.

Shapes parametrize over the zoo's distinct conv-shape classes
(SURVEY.md Appendix A): skinny-K expands, wide-K projects, SE FCs, 112²
wide-spatial depthwise through 7² channel-rich depthwise.
"""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

BF16 = torch.bfloat16
CL = torch.channels_last


@pytest.fixture(scope="module")
def ext():
    from yet_another_mobilenet_series_amd.ops.backend import ext as get_ext
    return get_ext()


def dev():
    return torch.device("cuda:0")


def randn_nhwc(n, c, h, w, seed=0, scale=1.0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    t = torch.randn(n, c, h, w, generator=g) * scale
    return t.to(dev()).to(BF16).contiguous(memory_format=CL)


def assert_close(got, want, rtol=1e-2, atol=None, msg=""):
    got = got.float().cpu()
    want = want.float().cpu()
    if atol is None:
        atol = 1e-2 * want.abs().mean().clamp(min=1e-3).item()
    torch.testing.assert_close(got, want, rtol=rtol, atol=atol, msg=msg)


# ---------------------------------------------------------------- pw GEMM

@pytest.mark.parametrize("m,k,n", [
    (96, 16, 64), (256, 672, 168), (1000, 72, 24), (12544, 160, 960),
    (4096, 960, 160), (256, 1280, 1000), (98, 8, 16),
])
def test_pw_conv_matches_fp32(ext, m, k, n):
    x = torch.randn(m, k, device=dev()).to(BF16)
    w = torch.randn(n, k, device=dev()).to(BF16) * 0.1
    y = torch.empty(m, n, device=dev(), dtype=BF16)
    ext.pw_conv(x.view(m, k, 1, 1), w, y.view(m, n, 1, 1))
    want = (x.float() @ w.float().t()).to(BF16)
    assert_close(y, want)


def test_pw_conv_identity_transpose_detect(ext):
    """A = I with ASYMMETRIC B catches swapped output layouts."""
    k = 64
    x = torch.eye(k, device=dev()).to(BF16)
    w = torch.arange(k * k, device=dev()).float().view(k, k)
    w = (w / (k * k)).to(BF16)  # asymmetric
    y = torch.empty(k, k, device=dev(), dtype=BF16)
    ext.pw_conv(x.view(k, k, 1, 1), w, y.view(k, k, 1, 1))
    assert torch.equal(y.float().cpu(), w.float().t().cpu())


@pytest.mark.parametrize("m,n1,n2", [(512, 64, 32), (12544, 16, 96),
                                     (50176, 64, 24), (256, 960, 240)])
def test_pw_wgrad_matches_fp32(ext, m, n1, n2):
    a = torch.randn(m, n1, device=dev()).to(BF16)
    b = torch.randn(m, n2, device=dev()).to(BF16)
    dw = torch.empty(n1, n2, device=dev(), dtype=torch.float32)
    ext.pw_conv_wgrad(a.view(m, n1, 1, 1), b.view(m, n2, 1, 1), dw)
    want = a.float().t() @ b.float()
    assert_close(dw, want, rtol=1e-2, atol=1e-2 * max(m, 1) ** 0.5 * 0.05)


def test_pw_wgrad_deterministic(ext):
    m, n1, n2 = 30000, 48, 32
    a = torch.randn(m, n1, device=dev()).to(BF16)
    b = torch.randn(m, n2, device=dev()).to(BF16)
    outs = []
    for _ in range(2):
        dw = torch.empty(n1, n2, device=dev(), dtype=torch.float32)
        ext.pw_conv_wgrad(a.view(m, n1, 1, 1), b.view(m, n2, 1, 1), dw)
        outs.append(dw.cpu())
    assert torch.equal(outs[0], outs[1])


# ------------------------------------------------------------- depthwise

DW_SHAPES = [  # (n, c, h, k, s) — zoo extremes from SURVEY.md §2.5
    (2, 16, 112, 3, 1), (2, 64, 112, 3, 2), (2, 72, 56, 5, 2),
    (2, 120, 28, 5, 1), (2, 240, 28, 3, 2), (2, 672, 14, 5, 2),
    (2, 960, 8, 5, 1), (3, 8, 9, 3, 1),
]


@pytest.mark.parametrize("n,c,h,k,s", DW_SHAPES)
def test_dw_fwd(ext, n, c, h, k, s):
    p = k // 2
    x = randn_nhwc(n, c, h, h, seed=1)
    w = (torch.randn(k * k, c, device=dev()) * 0.3).to(BF16)
    ho = (h + 2 * p - k) // s + 1
    y = torch.empty(n, c, ho, ho, device=dev(), dtype=BF16,
                    memory_format=CL)
    ext.dw_conv_fwd(x, w, y, k, s, p)
    wref = w.float().t().reshape(c, 1, k, k)
    want = F.conv2d(x.float(), wref, stride=s, padding=p, groups=c)
    assert_close(y, want)


@pytest.mark.parametrize("n,c,h,k,s", DW_SHAPES)
def test_dw_dgrad(ext, n, c, h, k, s):
    p = k // 2
    ho = (h + 2 * p - k) // s + 1
    dy = randn_nhwc(n, c, ho, ho, seed=2)
    w = (torch.randn(k * k, c, device=dev()) * 0.3).to(BF16)
    dx = torch.empty(n, c, h, h, device=dev(), dtype=BF16, memory_format=CL)
    ext.dw_conv_dgrad(dy, w, dx, k, s, p)
    wref = w.float().t().reshape(c, 1, k, k)
    want = torch.nn.grad.conv2d_input(
        (n, c, h, h), wref, dy.float(), stride=s, padding=p, groups=c)
    assert_close(dx, want)


@pytest.mark.parametrize("n,c,h,k,s", DW_SHAPES)
def test_dw_wgrad(ext, n, c, h, k, s):
    p = k // 2
    ho = (h + 2 * p - k) // s + 1
    x = randn_nhwc(n, c, h, h, seed=3)
    dy = randn_nhwc(n, c, ho, ho, seed=4)
    dw = torch.empty(c, k * k, device=dev(), dtype=torch.float32)
    ext.dw_conv_wgrad(x, dy, dw, k, s, p)
    want = torch.nn.grad.conv2d_weight(
        x.float(), (c, 1, k, k), dy.float(), stride=s, padding=p, groups=c)
    want = want.reshape(c, k * k)
    assert_close(dw, want, atol=0.05 * (n * ho * ho) ** 0.5 * 0.1)


# ------------------------------------------------------------------ stem

def test_im2row_matches_unfold(ext):
    n, h = 2, 32
    x = randn_nhwc(n, 3, h, h, seed=5)
    ho = h // 2
    kp = 32  # 27 rounded to /8
    patches = torch.empty(n * ho * ho, kp, device=dev(), dtype=BF16)
    ext.im2row(x, patches, 3, 2, 1)
    want = torch.nn.functional.unfold(x.float(), 3, padding=1, stride=2)
    # unfold: [n, cin*k*k, L] with cin-major taps; ours is tap-major, cin fastest
    want = want.view(n, 3, 9, -1).permute(0, 3, 2, 1).reshape(n * ho * ho, 27)
    assert torch.equal(patches[:, :27].float().cpu(), want.cpu())
    assert patches[:, 27:].abs().sum().item() == 0


def test_stem_conv_via_gemm(ext):
    """stem conv == im2row + GEMM (checked against F.conv2d)."""
    import torch.nn.functional as F
    n, h, cout = 2, 32, 32
    x = randn_nhwc(n, 3, h, h, seed=6)
    wref = (torch.randn(cout, 3, 3, 3, device=dev()) * 0.2).to(BF16)
    ho = h // 2
    m = n * ho * ho
    patches = torch.empty(m, 32, device=dev(), dtype=BF16)
    ext.im2row(x, patches, 3, 2, 1)
    wp = torch.zeros(cout, 32, device=dev(), dtype=BF16)
    wp[:, :27] = wref.permute(0, 2, 3, 1).reshape(cout, 27)
    y = torch.empty(m, cout, device=dev(), dtype=BF16)
    ext.pw_conv(patches.view(m, 32, 1, 1), wp, y.view(m, cout, 1, 1))
    want = F.conv2d(x.float(), wref.float(), stride=2, padding=1)
    want = want.permute(0, 2, 3, 1).reshape(m, cout)
    assert_close(y, want)


# ---------------------------------------------------------------- BN+act

def _bn_setup(n=4, c=72, h=14, seed=8):
    z = randn_nhwc(n, c, h, h, seed=seed, scale=2.0)
    gamma = (torch.rand(c, device=dev()) + 0.5).float()
    beta = (torch.randn(c, device=dev()) * 0.2).float()
    return z, gamma, beta


def test_bn_stats(ext):
    z, gamma, beta = _bn_setup()
    c = z.shape[1]
    mean = torch.empty(c, device=dev(), dtype=torch.float32)
    invstd = torch.empty_like(mean)
    ab = torch.empty(2 * c, device=dev(), dtype=torch.float32)
    rm = torch.zeros(c, device=dev())
    rv = torch.ones(c, device=dev())
    ext.bn_stats(z, mean, invstd, ab, gamma, beta, rm, rv, 0.1, 1e-5)
    zf = z.float()
    want_mean = zf.mean(dim=(0, 2, 3))
    want_var = zf.var(dim=(0, 2, 3), unbiased=False)
    assert_close(mean, want_mean, rtol=1e-3, atol=1e-3)
    assert_close(invstd, (want_var + 1e-5).rsqrt(), rtol=1e-3, atol=1e-3)
    m = z.shape[0] * z.shape[2] * z.shape[3]
    assert_close(rm, 0.1 * want_mean, rtol=1e-3, atol=1e-4)
    assert_close(rv, 0.9 + 0.1 * want_var * m / (m - 1), rtol=1e-3, atol=1e-3)
    a = gamma * invstd
    assert_close(ab[:c], a, rtol=1e-4, atol=1e-5)
    assert_close(ab[c:], beta - mean * a, rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("act", [0, 1, 2, 3])
@pytest.mark.parametrize("residual", [False, True])
def test_bn_act_fwd(ext, act, residual):
    z, gamma, beta = _bn_setup()
    zf = z.float()
    mean = zf.mean(dim=(0, 2, 3))
    invstd = (zf.var(dim=(0, 2, 3), unbiased=False) + 1e-5).rsqrt()
    res = randn_nhwc(*z.shape, seed=9) if residual else None
    y = torch.empty_like(z)
    a = gamma * invstd
    ab = torch.cat([a, beta - mean * a]).contiguous()
    ext.bn_act_fwd(z, ab, act, y, res)
    v = (zf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
    v = gamma.view(1, -1, 1, 1) * v + beta.view(1, -1, 1, 1)
    want = [lambda t: t, F.relu, F.relu6, F.hardswish][act](v)
    if residual:
        want = want + res.float()
    assert_close(y, want, rtol=1e-2, atol=2e-2)


@pytest.mark.parametrize("act", [0, 2, 3])
@pytest.mark.parametrize("training", [True, False])
def test_bn_act_bwd(ext, act, training):
    torch.manual_seed(0)
    z, gamma, beta = _bn_setup(c=64)
    n, c, h, w = z.shape
    m = n * h * w
    dy = randn_nhwc(n, c, h, w, seed=10)
    zf = z.float()
    mean = zf.mean(dim=(0, 2, 3))
    invstd = (zf.var(dim=(0, 2, 3), unbiased=False) + 1e-5).rsqrt()
    dgamma = torch.empty(c, device=dev(), dtype=torch.float32)
    dbeta = torch.empty_like(dgamma)
    dz = torch.empty_like(z)
    ext.bn_act_bwd(dy, z, mean, invstd, gamma, beta, act, training,
                   dgamma, dbeta, dz)
    # autograd reference on the same bf16-rounded values
    zr = zf.clone().requires_grad_(True)
    mu = mean.view(1, -1, 1, 1)
    iv = invstd.view(1, -1, 1, 1)
    if training:
        mu_r = zr.mean(dim=(0, 2, 3), keepdim=True)
        var_r = zr.var(dim=(0, 2, 3), unbiased=False, keepdim=True)
        zhat = (zr - mu_r) * (var_r + 1e-5).rsqrt()
    else:
        zhat = (zr - mu) * iv
    gr = gamma.clone().requires_grad_(True)
    br = beta.clone().requires_grad_(True)
    v = gr.view(1, -1, 1, 1) * zhat + br.view(1, -1, 1, 1)
    out = [lambda t: t, F.relu, F.relu6, F.hardswish][act](v)
    out.backward(dy.float())
    assert_close(dgamma, gr.grad, rtol=2e-2, atol=0.05 * m ** 0.5 * 0.02)
    assert_close(dbeta, br.grad, rtol=2e-2, atol=0.05 * m ** 0.5 * 0.02)
    assert_close(dz, zr.grad, rtol=5e-2, atol=5e-3)


# ----------------------------------------------------------------- misc

@pytest.mark.parametrize("act", [1, 4])
def test_bias_act_fwd_bwd(ext, act):
    m, c = 37, 64
    z = torch.randn(m, c, device=dev()).to(BF16)
    bias = torch.randn(c, device=dev()).float()
    y = torch.empty_like(z)
    ext.bias_act_fwd(z, bias, act, y)
    v = z.float() + bias
    want = F.relu(v) if act == 1 else F.hardsigmoid(v)
    assert_close(y, want, rtol=1e-2, atol=1e-2)

    dy = torch.randn(m, c, device=dev()).to(BF16)
    dz = torch.empty_like(dy)
    dbias = torch.empty(c, device=dev(), dtype=torch.float32)
    ext.bias_act_bwd(dy, z, bias, act, dz, dbias)
    vr = z.float().clone().requires_grad_(True)
    br = bias.clone().requires_grad_(True)
    out = F.relu(vr + br) if act == 1 else F.hardsigmoid(vr + br)
    out.backward(dy.float())
    assert_close(dz, vr.grad, rtol=1e-2, atol=1e-3)
    assert_close(dbias, br.grad, rtol=1e-2, atol=1e-2)


def test_rowscale_fwd_bwd(ext):
    n, c, h = 3, 48, 14
    x = randn_nhwc(n, c, h, h, seed=11)
    s = torch.rand(n, c, device=dev()).to(BF16)
    y = torch.empty_like(x)
    ext.rowscale_fwd(x, s, y)
    want = x.float() * s.float().view(n, c, 1, 1)
    assert_close(y, want)

    dy = randn_nhwc(n, c, h, h, seed=12)
    dx = torch.empty_like(x)
    ds = torch.empty(n, c, device=dev(), dtype=BF16)
    ext.rowscale_bwd(dy, x, s, dx, ds)
    assert_close(dx, dy.float() * s.float().view(n, c, 1, 1))
    assert_close(ds, (dy.float() * x.float()).sum(dim=(2, 3)),
                 rtol=2e-2, atol=0.1)


def test_avgpool_fwd_bwd(ext):
    n, c, h = 3, 96, 7
    x = randn_nhwc(n, c, h, h, seed=13)
    out = torch.empty(n, c, device=dev(), dtype=BF16)
    ext.avgpool_fwd(x, out)
    assert_close(out, x.float().mean(dim=(2, 3)), rtol=1e-2, atol=1e-2)

    dp = torch.randn(n, c, device=dev()).to(BF16)
    dx = torch.empty_like(x)
    ext.avgpool_bwd(dp, dx)
    want = (dp.float() / (h * h)).view(n, c, 1, 1).expand(n, c, h, h)
    assert_close(dx, want)
    dx2 = dx.clone()
    ext.avgpool_bwd_add(dp, dx2)
    assert_close(dx2, 2 * want, rtol=1e-2, atol=1e-3)


def test_label_smooth_ce(ext):
    n, c = 16, 1000
    logits = (torch.randn(n, c, device=dev()) * 3).to(BF16)
    target = torch.randint(0, c, (n,), device=dev())
    loss = torch.empty((), device=dev(), dtype=torch.float32)
    dlogits = torch.empty(n, c, device=dev(), dtype=torch.float32)
    ext.label_smooth_ce(logits, target, 0.1, loss, dlogits)
    lf = logits.float()
    logp = F.log_softmax(lf, dim=1)
    nll = -logp.gather(1, target.view(-1, 1)).squeeze(1)
    smooth = -logp.mean(dim=1)
    want = (0.9 * nll + 0.1 * smooth).mean()
    assert_close(loss, want, rtol=1e-3, atol=1e-3)
    lr = lf.clone().requires_grad_(True)
    logp = F.log_softmax(lr, dim=1)
    ((0.9 * (-logp.gather(1, target.view(-1, 1)).squeeze(1))
      + 0.1 * (-logp.mean(dim=1))).mean()).backward()
    assert_close(dlogits, lr.grad, rtol=1e-3, atol=1e-5)


# ------------------------------------------------------- fused updates

def test_fused_sgd_matches_torch(ext):
    torch.manual_seed(0)
    shapes = [(64, 32), (128,), (16, 8, 3, 3), (1000, 1280)]
    ps1 = [torch.randn(*s_, device=dev()) for s_ in shapes]
    gs = [torch.randn(*s_, device=dev()) for s_ in shapes]
    ps2 = [p.clone() for p in ps1]
    m1 = [torch.zeros_like(p) for p in ps1]
    for _ in range(3):
        ext.fused_sgd(ps1, gs, m1, [1e-4, 0.0, 1e-4, 1e-4], 0.1, 0.9, True)
    opt = torch.optim.SGD([
        {"params": [ps2[0], ps2[2], ps2[3]], "weight_decay": 1e-4},
        {"params": [ps2[1]], "weight_decay": 0.0}], lr=0.1, momentum=0.9,
        nesterov=True)
    for _ in range(3):
        for p, g in zip(ps2, gs):
            p.grad = g.clone()
        opt.step()
    for a, b in zip(ps1, ps2):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_fused_rmsprop_tf_matches_cpu_impl(ext):
    from yet_another_mobilenet_series_amd.utils.optim import RMSpropTF
    torch.manual_seed(1)
    shapes = [(48, 24), (96,)]
    ps1 = [torch.randn(*s_, device=dev()) for s_ in shapes]
    gs = [torch.randn(*s_, device=dev()) for s_ in shapes]
    ps2 = [torch.nn.Parameter(p.clone()) for p in ps1]
    ms = [torch.ones_like(p) for p in ps1]
    mom = [torch.zeros_like(p) for p in ps1]
    for _ in range(3):
        ext.fused_rmsprop_tf(ps1, gs, ms, mom, [1e-5, 0.0], 0.064, 0.9,
                             1e-3, 0.9)
    opt = RMSpropTF([{"params": [ps2[0]], "weight_decay": 1e-5},
                     {"params": [ps2[1]], "weight_decay": 0.0}],
                    lr=0.064, alpha=0.9, eps=1e-3, momentum=0.9)
    for _ in range(3):
        for p, g in zip(ps2, gs):
            p.grad = g.clone()
        opt.step()
    for a, b in zip(ps1, ps2):
        torch.testing.assert_close(a, b.detach(), rtol=1e-5, atol=1e-6)


def test_fused_ema_matches_reference(ext):
    torch.manual_seed(2)
    vals = [torch.randn(100, device=dev()), torch.randn(7, 9, device=dev())]
    sh1 = [torch.randn_like(v) for v in vals]
    sh2 = [s_.clone() for s_ in sh1]
    ext.fused_ema(vals, sh1, 0.99)
    for v, s_ in zip(vals, sh2):
        s_.mul_(0.99).add_(v, alpha=0.01)
    for a, b in zip(sh1, sh2):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)


def test_dispatch_fallthrough_hard_errors(ext):
    """An unsupported (k, stride, lop, act) combo must raise, never return
    uninitialized output (VERDICT round-1 weak #6)."""
    x = randn_nhwc(2, 16, 8, 8)
    w = torch.zeros(49, 16, dtype=BF16, device=dev())
    y = torch.empty_like(x)
    with pytest.raises(RuntimeError, match="no kernel instantiation"):
        ext.dw_conv_fwd(x, w, y, 7, 1, 3)  # k=7 not in the zoo
    w3 = torch.zeros(9, 16, dtype=BF16, device=dev())
    coef = torch.ones(32, dtype=torch.float32, device=dev())
    with pytest.raises(RuntimeError, match="no kernel instantiation"):
        # bnact load-op with identity act has no instantiation
        ext.dw_conv_fwd(x, w3, y, 3, 1, 1, lop_kind=1, lop_act=0,
                        lop_coef=coef)


def test_forward_changes_after_fused_optimizer_step():
    """ADVICE round-1 HIGH regression test: after a FusedSGD step the next
    forward must see the UPDATED conv/linear weights (the bf16 pack cache
    must be invalidated by the raw-pointer fused update)."""
    from yet_another_mobilenet_series_amd import models
    from yet_another_mobilenet_series_amd.ops import functional as OF
    from yet_another_mobilenet_series_amd.utils.optim import FusedSGD

    torch.manual_seed(0)
    model = models.build_model("mobilenet_v2", dropout=0.0,
                               width_mult=0.35).to(dev())
    model = model.to(memory_format=torch.channels_last)
    model.train()
    x = randn_nhwc(4, 3, 64, 64, seed=3)
    y = torch.randint(0, 1000, (4,), device=dev())
    opt = FusedSGD(model.parameters(), lr=0.5, momentum=0.9)

    logits0 = model(x).detach().float()
    loss = OF.label_smooth_ce(model(x), y, 0.1)
    opt.zero_grad(set_to_none=False)
    loss.backward()
    opt.step()
    model.eval()  # freeze BN stats so only the weight change shows
    with torch.no_grad():
        la = model(x).detach().float()
        opt.zero_grad(set_to_none=False)
    model.train()
    loss2 = OF.label_smooth_ce(model(x), y, 0.1)
    loss2.backward()
    opt.step()
    model.eval()
    with torch.no_grad():
        lb = model(x).detach().float()
    # conv/linear weights moved between the two eval forwards -> logits move
    assert (la - lb).abs().max().item() > 1e-3, \
        "forward unchanged after optimizer step: stale weight-pack cache"


# ---- fused squeeze-excite (K7, se.hip) ------------------------------------

def _se_ref(x32, w1, b1, w2, b2):
    """fp32 torch reference of the SE chain from bf16-rounded x."""
    pooled = x32.mean(dim=(2, 3))                       # [N, C]
    ha = torch.relu(pooled @ w1.t() + b1)               # [N, R]
    sz = ha @ w2.t() + b2                               # [N, C]
    s = torch.clamp(sz + 3, 0, 6) / 6
    y = x32 * s[:, :, None, None]
    return y, pooled, ha, sz, s


@pytest.mark.parametrize("c,r,hw", [(16, 8, 56), (72, 24, 28), (120, 32, 28),
                                    (480, 120, 14), (672, 168, 7),
                                    (960, 240, 7)])
def test_se_fwd_fused(ext, c, r, hw):
    torch.manual_seed(c + r)
    x = randn_nhwc(5, c, hw, hw, seed=c)
    w1 = (torch.randn(r, c) / c ** 0.5).to(dev())
    b1 = torch.randn(r, device=dev()) * 0.1
    w2 = (torch.randn(c, r) / r ** 0.5).to(dev())
    b2 = torch.randn(c, device=dev()) * 0.1
    y = torch.empty_like(x)
    pooled = torch.empty(5, c, dtype=torch.float32, device=dev())
    ha = torch.empty(5, r, dtype=torch.float32, device=dev())
    sz = torch.empty(5, c, dtype=torch.float32, device=dev())
    ext.se_fwd(x, w1.bfloat16().t().contiguous(), b1,
               w2.bfloat16().t().contiguous(), b2, y, pooled, ha, sz)
    # reference from the same bf16-rounded operands
    yr, pr, har, szr, _ = _se_ref(x.float(), w1.bfloat16().float(), b1,
                                  w2.bfloat16().float(), b2)
    assert_close(pooled, pr, rtol=2e-3)
    assert_close(ha, har, rtol=2e-3)
    assert_close(sz, szr, rtol=5e-3)
    assert_close(y, yr, rtol=1e-2)


def test_se_fwd_with_bnact_lop(ext):
    """LOP_BNACT path: BN+act applied on load, activated input streamed to
    ymat, SE math on the activated values."""
    c, r, hw = 120, 32, 14
    z = randn_nhwc(4, c, hw, hw, seed=9)
    ab = torch.randn(2 * c, device=dev()).contiguous()
    ab[:c] = ab[:c].abs() + 0.5
    w1 = (torch.randn(r, c) / c ** 0.5).to(dev()).bfloat16().contiguous()
    b1 = torch.randn(r, device=dev()) * 0.1
    w2 = (torch.randn(c, r) / r ** 0.5).to(dev()).bfloat16().contiguous()
    b2 = torch.randn(c, device=dev()) * 0.1
    y = torch.empty_like(z)
    ymat = torch.empty_like(z)
    pooled = torch.empty(4, c, dtype=torch.float32, device=dev())
    ha = torch.empty(4, r, dtype=torch.float32, device=dev())
    sz = torch.empty(4, c, dtype=torch.float32, device=dev())
    ext.se_fwd(z, w1.t().contiguous(), b1, w2.t().contiguous(), b2, y,
               pooled, ha, sz, lop_kind=1, lop_act=3, lop_coef=ab,
               ymat=ymat)
    zv = z.float()
    v = zv * ab[:c].view(1, c, 1, 1) + ab[c:].view(1, c, 1, 1)
    act = v * torch.clamp(v + 3, 0, 6) / 6  # hswish
    act = act.bfloat16().float()            # ymat is bf16-rounded
    assert_close(ymat, act, rtol=1e-2)
    yr, pr, _, _, _ = _se_ref(act, w1.float(), b1, w2.float(), b2)
    assert_close(pooled, pr, rtol=3e-3)
    assert_close(y, yr, rtol=1e-2)


@pytest.mark.parametrize("c,r,hw", [(16, 8, 56), (480, 120, 14),
                                    (960, 240, 7)])
def test_se_bwd_fused(ext, c, r, hw):
    """Full SE backward vs autograd on the fp32 reference (same bf16
    inputs): the fused kernels must carry NO extra rounding noise."""
    n = 6
    torch.manual_seed(3)
    x = randn_nhwc(n, c, hw, hw, seed=c + 1)
    dy = randn_nhwc(n, c, hw, hw, seed=c + 2)
    w1 = ((torch.randn(r, c) / c ** 0.5).to(dev()).bfloat16())
    b1 = torch.randn(r, device=dev()) * 0.1
    w2 = ((torch.randn(c, r) / r ** 0.5).to(dev()).bfloat16())
    b2 = torch.randn(c, device=dev()) * 0.1

    # fused path
    y = torch.empty_like(x)
    pooled = torch.empty(n, c, dtype=torch.float32, device=dev())
    ha = torch.empty(n, r, dtype=torch.float32, device=dev())
    sz = torch.empty(n, c, dtype=torch.float32, device=dev())
    ext.se_fwd(x, w1.t().contiguous(), b1, w2.t().contiguous(), b2, y,
               pooled, ha, sz)
    dx = torch.empty_like(x)
    dsz = torch.empty(n, c, dtype=torch.float32, device=dev())
    dhz = torch.empty(n, r, dtype=torch.float32, device=dev())
    ext.se_bwd(dy, x, w1.contiguous(), w2.contiguous(), ha, sz, dx, dsz,
               dhz)
    dw1 = torch.empty(r, c, dtype=torch.float32, device=dev())
    dw2 = torch.empty(c, r, dtype=torch.float32, device=dev())
    db1 = torch.empty(r, dtype=torch.float32, device=dev())
    db2 = torch.empty(c, dtype=torch.float32, device=dev())
    ext.se_wgrad(dsz, ha, dhz, pooled, dw1, dw2, db1, db2)

    # fp32 autograd reference from the same bf16-rounded operands
    xr = x.float().requires_grad_(True)
    w1r = w1.float().requires_grad_(True)
    b1r = b1.detach().clone().requires_grad_(True)
    w2r = w2.float().requires_grad_(True)
    b2r = b2.detach().clone().requires_grad_(True)
    yr, _, _, _, _ = _se_ref(xr, w1r, b1r, w2r, b2r)
    yr.backward(dy.float())
    assert_close(dx, xr.grad, rtol=2e-2)
    assert_close(dw1, w1r.grad, rtol=1e-2)
    assert_close(dw2, w2r.grad, rtol=1e-2)
    assert_close(db1, b1r.grad, rtol=1e-2)
    assert_close(db2, b2r.grad, rtol=1e-2)


# ---- dropout (K14) --------------------------------------------------------

def test_dropout_kernel(ext):
    torch.manual_seed(11)
    x = torch.randn(64, 1280, device=dev()).bfloat16().contiguous()
    y = torch.empty_like(x)
    p = 0.2
    ext.dropout(x, y, 12345, p, forward=True)
    scale = 1.0 / (1.0 - p)
    zero = (y == 0)
    frac = zero.float().mean().item()
    assert abs(frac - p) < 0.02, frac  # ~Bernoulli(p) at 82k samples
    kept = ~zero
    assert torch.allclose(y[kept].float(), (x[kept].float() * scale)
                          .bfloat16().float())
    # backward regenerates the SAME mask from the seed
    dy = torch.randn_like(x.float()).bfloat16()
    dx = torch.empty_like(dy)
    ext.dropout(dy, dx, 12345, p, forward=False)
    assert (dx[zero] == 0).all()
    assert torch.allclose(dx[kept].float(), (dy[kept].float() * scale)
                          .bfloat16().float())
    # a different seed gives a different mask
    y2 = torch.empty_like(x)
    ext.dropout(x, y2, 54321, p, forward=True)
    assert not torch.equal(y, y2)


def test_dropout_autograd_path():
    from yet_another_mobilenet_series_amd.ops import functional as OF
    torch.manual_seed(0)
    x = torch.randn(32, 256, device=dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    y = OF.dropout(x, 0.5, training=True)
    dy = torch.ones_like(y)
    y.backward(dy)
    # grad is nonzero exactly where the output was kept
    assert torch.equal((x.grad != 0), (y != 0).detach())
    # eval mode: identity
    assert OF.dropout(x, 0.5, training=False) is x


def test_fused_repack_matches_python_packs(ext):
    """After the optimizer mutates a param, refresh_packs must leave every
    cached pack equal to a from-scratch Python repack."""
    from yet_another_mobilenet_series_amd.ops.hip_ops import (
        _packed, refresh_packs)
    torch.manual_seed(2)
    pw = torch.nn.Parameter(torch.randn(64, 32, 1, 1, device=dev()))
    dw = torch.nn.Parameter(torch.randn(48, 1, 5, 5, device=dev()))
    # build the caches
    p1 = _packed(pw, "pw", lambda w: w.view(64, 32).to(BF16).contiguous(),
                 spec="cast")
    p2 = _packed(pw, "pw_t", lambda w: w.view(64, 32).t().to(BF16)
                 .contiguous(), spec="t")
    p3 = _packed(dw, "dw", lambda w: w.view(48, 25).t().to(BF16)
                 .contiguous(), spec="t")
    with torch.no_grad():
        pw.mul_(1.5)
        dw.add_(0.25)
    refresh_packs([pw, dw])
    torch.cuda.synchronize()
    assert torch.equal(p1, pw.detach().view(64, 32).to(BF16))
    assert torch.equal(p2, pw.detach().view(64, 32).t().to(BF16)
                       .contiguous())
    assert torch.equal(p3, dw.detach().view(48, 25).t().to(BF16)
                       .contiguous())
    # entries re-stamped: next _packed call is a cache HIT (same object)
    q1 = _packed(pw, "pw", lambda w: w.view(64, 32).to(BF16).contiguous(),
                 spec="cast")
    assert q1 is p1


def test_pw_conv_residual_epilogue(ext):
    """C = A @ B^T + res fused in the GEMM epilogue (backward residual)."""
    m, k_, n = 512, 64, 32
    a = torch.randn(m, k_, device=dev()).bfloat16().contiguous()
    b = torch.randn(n, k_, device=dev()).bfloat16().contiguous()
    res = torch.randn(m, n, device=dev()).bfloat16().contiguous()
    y = torch.empty(m, n, device=dev(), dtype=BF16)
    ext.pw_conv(a.view(m, k_, 1, 1), b, y.view(m, n, 1, 1),
                residual=res.view(m, n, 1, 1))
    want = (a.float() @ b.float().t() + res.float())
    assert_close(y, want)


def test_guard_bands_model_step(ext):
    """OOB sweep: a full fused V3-Small fwd+bwd+optimizer
    step with EVERY Python-level output allocation wrapped in canary
    guard bands — any HIP kernel writing outside its buffers trips it."""
    from yet_another_mobilenet_series_amd.models import build_model
    from yet_another_mobilenet_series_amd.ops.guard import guard_allocations
    from yet_another_mobilenet_series_amd.utils.optim import FusedSGD

    torch.manual_seed(0)
    model = build_model("mobilenet_v3_small").to(dev()) \
        .to(memory_format=CL).train()
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9,
                   weight_decay=1e-5)
    x = torch.randn(32, 3, 224, 224, device=dev()) \
        .to(BF16).contiguous(memory_format=CL)
    tgt = torch.randint(0, 1000, (32,), device=dev())
    with guard_allocations() as chk:
        for _ in range(2):
            opt.zero_grad(set_to_none=True)
            loss = F.cross_entropy(model(x).float(), tgt)
            loss.backward()
            opt.step()
        chk()
    assert torch.isfinite(loss.detach()).item()

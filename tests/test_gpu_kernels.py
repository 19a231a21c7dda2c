"""HIP kernel numerics on MI355X — every kernel vs the plain-PyTorch
fp32 oracle defined by the CPU ops."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from bdbnn_amd import _C
from bdbnn_amd.ops.binarize import binsign, weight_scale
from bdbnn_amd.ops.binary_conv import BinaryConvFunction, HardBinaryConv
from bdbnn_amd.ops.kurtosis import kurtosis_loss_fused
from bdbnn_amd.ops.optim import FusedSGD, FusedAdam


def _nat():
    return _C.native_required()


def _cl(x):
    return x.contiguous(memory_format=torch.channels_last)


# ---------------- pack / decode ----------------

@pytest.mark.parametrize("C", [32, 64, 48, 96])
def test_sign_pack_bits(C):
    torch.manual_seed(0)
    x = torch.randn(2, C, 5, 7, device="cuda")
    xp = _nat().sign_pack_nhwc(_cl(x))
    assert xp.shape == (2, 5, 7, (C + 31) // 32)
    bits = xp.cpu().numpy().astype("uint32")
    xs = x.permute(0, 2, 3, 1).cpu()  # NHWC
    for n in range(2):
        for h in range(5):
            for w in range(7):
                for c in range(C):
                    want = 1 if xs[n, h, w, c].item() >= 0 else 0
                    got = (int(bits[n, h, w, c // 32]) >> (c % 32)) & 1
                    assert got == want
                # garbage bits of the a-pack are 0
                tail = C % 32
                if tail:
                    assert (int(bits[n, h, w, C // 32]) >> tail) == 0


def test_binsign_decode():
    x = torch.randn(3, 16, 4, 4, device="cuda")
    y = _nat().binsign_decode(_cl(x), False)
    assert torch.equal(y, _cl(binsign(x)))
    yb = _nat().binsign_decode(_cl(x), True)
    assert yb.dtype == torch.bfloat16
    assert torch.equal(yb.float(), _cl(binsign(x)))


def test_ste_mask_modes():
    g = torch.randn(4, 8, 6, 6, device="cuda")
    x = torch.randn(4, 8, 6, 6, device="cuda") * 2
    out = _nat().ste_mask_mul(_cl(g), _cl(x), 0, 0.0, 0.0)
    ref = _cl(g * (x.abs() <= 1).float())
    assert torch.allclose(out, ref)
    out = _nat().ste_mask_mul(_cl(g), _cl(x), 1, 0.0, 0.0)
    neg = (x >= -1) & (x < 0)
    pos = (x >= 0) & (x < 1)
    m = torch.where(neg, 2 + 2 * x, torch.where(pos, 2 - 2 * x,
                                                torch.zeros_like(x)))
    assert torch.allclose(out, _cl(g * m), atol=1e-6)
    out = _nat().ste_mask_mul(_cl(g), _cl(x), 2, 2.0, 1.5)
    th = torch.tanh(2.0 * x)
    assert torch.allclose(out, _cl(g * 1.5 * 2.0 * (1 - th * th)), atol=1e-5)


# ---------------- xnor conv ----------------

@pytest.mark.parametrize("shape", [
    # (N, C, H, W, K, ksize, stride, pad)
    (2, 64, 14, 14, 64, 3, 1, 1),
    (2, 128, 9, 11, 96, 3, 1, 1),     # non-multiple-of-tile K, odd spatial
    (1, 64, 8, 8, 128, 1, 2, 0),      # 1x1 stride-2 downsample
    (2, 48, 10, 10, 64, 3, 2, 1),     # tail channel word + stride 2
    (1, 512, 7, 7, 512, 3, 1, 1),     # deepest layer shape
    (3, 96, 6, 6, 16, 3, 1, 1),       # small K
    (2, 16, 8, 8, 16, 3, 1, 1),       # CIFAR stage-1 (C=16, tail word)
])
def test_xnor_conv_matches_fp32_conv(shape):
    N, C, H, W, K, ks, stride, pad = shape
    torch.manual_seed(1)
    x = torch.randn(N, C, H, W, device="cuda")
    w = torch.randn(K, C, ks, ks, device="cuda") * 0.5
    nat = _nat()
    xp = nat.sign_pack_nhwc(_cl(x))
    wp, alpha, stab = nat.weight_pack(w)
    out = nat.xnor_conv_fwd(xp, wp, alpha, stab, C, stride, pad, False,
                            False)[0]
    ref = F.conv2d(binsign(x), weight_scale(w) * binsign(w), None,
                   stride=stride, padding=pad)
    # integer dot + fp32 scale: exact up to fp32 rounding of alpha*int
    assert out.shape == ref.shape
    assert torch.allclose(_cl(out), _cl(ref), atol=1e-3, rtol=1e-4), \
        (out - ref).abs().max().item()


def test_xnor_conv_bf16_out():
    x = torch.randn(2, 64, 8, 8, device="cuda")
    w = torch.randn(32, 64, 3, 3, device="cuda")
    nat = _nat()
    xp = nat.sign_pack_nhwc(_cl(x))
    wp, alpha, stab = nat.weight_pack(w)
    out = nat.xnor_conv_fwd(xp, wp, alpha, stab, 64, 1, 1, True, False)[0]
    assert out.dtype == torch.bfloat16
    ref = F.conv2d(binsign(x), weight_scale(w) * binsign(w), None, 1, 1)
    assert torch.allclose(out.float(), ref, atol=0.05, rtol=0.02)


def test_binary_conv_autograd_gpu_vs_cpu():
    torch.manual_seed(3)
    x = (torch.randn(2, 64, 10, 10) * 1.5)
    w = torch.randn(32, 64, 3, 3) * 1.2

    xg = _cl(x.cuda()).requires_grad_(True)
    wg = w.cuda().requires_grad_(True)
    out_g, _, _ = BinaryConvFunction.apply(xg, wg, 1, 1, "ste", None, None)
    g = torch.randn_like(out_g)
    out_g.backward(g)

    xc = x.clone().requires_grad_(True)
    wc = w.clone().requires_grad_(True)
    out_c, _, _ = BinaryConvFunction.apply(xc, wc, 1, 1, "ste", None, None)
    out_c.backward(g.cpu())

    assert torch.allclose(out_g.cpu(), out_c, atol=1e-3, rtol=1e-4)
    assert torch.allclose(xg.grad.cpu(), xc.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(wg.grad.cpu(), wc.grad, atol=1e-3, rtol=1e-3)


def test_hard_binary_conv_module_gpu():
    conv = HardBinaryConv(64, 64, 3, 1, 1).cuda()
    conv = conv.to(memory_format=torch.channels_last)
    x = _cl(torch.randn(2, 64, 14, 14, device="cuda")).requires_grad_(True)
    out = conv(x)
    out.sum().backward()
    assert conv.weight.grad is not None and x.grad is not None
    assert torch.isfinite(out).all()


# ---------------- kurtosis ----------------

def test_kurtosis_fused_gpu_matches_cpu():
    torch.manual_seed(4)
    ws_cpu = [torch.randn(64, 64, 3, 3) for _ in range(3)]
    tgts = [1.8, 1.4, 1.2]
    loss_c, kurts_c = kurtosis_loss_fused(
        [w.clone().requires_grad_(True) for w in ws_cpu], tgts, "sum")
    ws_gpu = [w.cuda().requires_grad_(True) for w in ws_cpu]
    loss_g, kurts_g = kurtosis_loss_fused(ws_gpu, tgts, "sum")
    assert torch.allclose(loss_g.cpu(), loss_c, atol=1e-4, rtol=1e-4)
    assert torch.allclose(kurts_g.cpu(), kurts_c, atol=1e-4, rtol=1e-4)
    loss_g.backward()
    ws_ref = [w.clone().requires_grad_(True) for w in ws_cpu]
    loss_ref, _ = kurtosis_loss_fused(ws_ref, tgts, "sum")
    loss_ref.backward()
    for g_gpu, g_cpu in zip(ws_gpu, ws_ref):
        assert torch.allclose(g_gpu.grad.cpu(), g_cpu.grad,
                              atol=1e-5, rtol=1e-4)


# ---------------- weight KD ----------------

def test_weight_kd_fused_gpu():
    from bdbnn_amd.ops.kd import _FusedWeightKD
    torch.manual_seed(5)
    ws = [torch.randn(32, 16, 3, 3, device="cuda").requires_grad_(True)
          for _ in range(2)]
    wt = [torch.randn(32, 16, 3, 3, device="cuda") for _ in range(2)]
    loss = _FusedWeightKD.apply(2, *ws, *wt)
    ref = sum((torch.exp(b) * (b - a)).mean() for a, b in zip(ws, wt))
    assert torch.allclose(loss, ref, atol=1e-4, rtol=1e-4)
    loss.backward()
    for a, b in zip(ws, wt):
        assert torch.allclose(a.grad, -torch.exp(b) / b.numel(),
                              atol=1e-6, rtol=1e-5)


# ---------------- fused optimizers ----------------

def test_fused_sgd_matches_torch():
    torch.manual_seed(6)
    ps = [torch.randn(100, device="cuda").requires_grad_(True),
          torch.randn(17, 3, 3, 3, device="cuda").requires_grad_(True)]
    ref = [p.detach().clone().requires_grad_(True) for p in ps]
    for p, r in zip(ps, ref):
        g = torch.randn_like(p)
        p.grad = g.clone()
        r.grad = g.clone()
    opt = FusedSGD(ps, lr=0.1, momentum=0.9, weight_decay=1e-4)
    topt = torch.optim.SGD(ref, lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(3):
        opt.step()
        topt.step()
        for p in ps + ref:
            p.grad = p.grad * 0.9 + 0.01  # evolve grads deterministically
    for p, r in zip(ps, ref):
        assert torch.allclose(p, r, atol=1e-5, rtol=1e-5)


def test_fused_adam_matches_torch():
    torch.manual_seed(7)
    ps = [torch.randn(257, device="cuda").requires_grad_(True)]
    ref = [p.detach().clone().requires_grad_(True) for p in ps]
    for p, r in zip(ps, ref):
        g = torch.randn_like(p)
        p.grad = g.clone()
        r.grad = g.clone()
    opt = FusedAdam(ps, lr=1e-3, weight_decay=1e-4)
    topt = torch.optim.Adam(ref, lr=1e-3, weight_decay=1e-4)
    for _ in range(3):
        opt.step()
        topt.step()
    for p, r in zip(ps, ref):
        assert torch.allclose(p, r, atol=1e-6, rtol=1e-5)


# ---------------- end-to-end ----------------

def test_resnet18_gpu_step():
    from bdbnn_amd.models import imagenet as im
    m = im.resnet18(False).cuda().to(memory_format=torch.channels_last)
    x = _cl(torch.randn(4, 3, 64, 64, device="cuda"))
    y = torch.randint(0, 1000, (4,), device="cuda")
    out = m(x)
    loss = torch.nn.functional.cross_entropy(out, y)
    loss.backward()
    assert torch.isfinite(loss)
    grads_ok = [p.grad is not None for p in m.parameters()]
    assert all(grads_ok)


# ---------------- fused PReLU ----------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_channel_prelu_fused(dtype):
    from bdbnn_amd.ops.activations import ChannelPReLU
    torch.manual_seed(8)
    m = ChannelPReLU(32).cuda()
    with torch.no_grad():
        m.weight.uniform_(-0.5, 0.5)
    x = (torch.randn(4, 32, 9, 9, device="cuda", dtype=dtype) * 2)
    x = _cl(x).requires_grad_(True)
    out = m(x)
    g = torch.randn_like(out)
    out.backward(g)

    ref_m = torch.nn.PReLU(32).cuda()
    with torch.no_grad():
        ref_m.weight.copy_(m.weight)
    x2 = x.detach().clone().requires_grad_(True)
    out2 = ref_m(x2.float())
    out2.backward(g.float())

    atol = 1e-5 if dtype == torch.float32 else 0.05
    assert torch.allclose(out.float(), out2, atol=atol, rtol=1e-2)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=atol,
                          rtol=1e-2)
    assert torch.allclose(m.weight.grad, ref_m.weight.grad,
                          atol=0.05 if dtype == torch.bfloat16 else 1e-3,
                          rtol=1e-2)


# ---------------- fused BN (+add) (+act) ----------------

@pytest.mark.parametrize("act_kind", ["prelu", "relu", "none"])
@pytest.mark.parametrize("with_skip", [False, True])
def test_fused_bn_act_train(act_kind, with_skip):
    from bdbnn_amd.ops.bn_act import fused_bn_act
    from bdbnn_amd.ops.activations import ChannelPReLU
    torch.manual_seed(10)
    C = 32
    bn = torch.nn.BatchNorm2d(C).cuda()
    bn2 = torch.nn.BatchNorm2d(C).cuda()
    with torch.no_grad():
        bn.weight.uniform_(0.5, 1.5); bn.bias.uniform_(-0.3, 0.3)
        bn2.load_state_dict(bn.state_dict())
    act = act2 = None
    if act_kind == "prelu":
        act = ChannelPReLU(C).cuda()
        act2 = torch.nn.PReLU(C).cuda()
        with torch.no_grad():
            act.weight.uniform_(-0.4, 0.6)
            act2.weight.copy_(act.weight)
    elif act_kind == "relu":
        act = act2 = "relu"
    x = _cl(torch.randn(8, C, 14, 14, device="cuda") * 2).requires_grad_(True)
    skip = (_cl(torch.randn(8, C, 14, 14, device="cuda")).requires_grad_(True)
            if with_skip else None)
    bn.train(); bn2.train()
    out = fused_bn_act(x, bn, act, skip=skip)

    # reference composition
    x2 = x.detach().clone().requires_grad_(True)
    skip2 = skip.detach().clone().requires_grad_(True) if with_skip else None
    z = bn2(x2)
    if with_skip:
        z = z + skip2
    if act_kind == "prelu":
        ref = act2(z)
    elif act_kind == "relu":
        ref = torch.relu(z)
    else:
        ref = z
    assert torch.allclose(out, _cl(ref), atol=2e-4, rtol=1e-4), \
        (out - _cl(ref)).abs().max().item()

    g = torch.randn_like(out)
    out.backward(g)
    ref.backward(_cl(g))
    assert torch.allclose(x.grad, x2.grad, atol=2e-4, rtol=1e-3)
    if with_skip:
        assert torch.allclose(skip.grad, skip2.grad, atol=2e-4, rtol=1e-3)
    assert torch.allclose(bn.weight.grad, bn2.weight.grad, atol=2e-3,
                          rtol=1e-3)
    assert torch.allclose(bn.bias.grad, bn2.bias.grad, atol=2e-3, rtol=1e-3)
    if act_kind == "prelu":
        assert torch.allclose(act.weight.grad, act2.weight.grad, atol=2e-3,
                              rtol=1e-3)
    # running stats updated identically
    assert torch.allclose(bn.running_mean, bn2.running_mean, atol=1e-4)
    assert torch.allclose(bn.running_var, bn2.running_var, atol=1e-4)
    assert bn.num_batches_tracked.item() == bn2.num_batches_tracked.item()


def test_fused_bn_act_eval():
    from bdbnn_amd.ops.bn_act import fused_bn_act
    torch.manual_seed(11)
    C = 16
    bn = torch.nn.BatchNorm2d(C).cuda()
    with torch.no_grad():
        bn.running_mean.uniform_(-1, 1)
        bn.running_var.uniform_(0.5, 2)
        bn.weight.uniform_(0.5, 1.5)
    bn.eval()
    x = _cl(torch.randn(4, C, 7, 7, device="cuda"))
    with torch.no_grad():
        out = fused_bn_act(x, bn, "relu")
        ref = torch.relu(bn(x))
    assert torch.allclose(out, _cl(ref), atol=1e-4, rtol=1e-4)


# ---------------- inference path ----------------

def test_packed_inference_matches_module_forward():
    from bdbnn_amd.models import imagenet as im
    from bdbnn_amd.engine import PackedInference
    torch.manual_seed(12)
    model = im.resnet18(False)
    ref = im.resnet18(False)
    ref.load_state_dict(model.state_dict())
    ref = ref.cuda().to(memory_format=torch.channels_last).eval()
    eng = PackedInference(model, dtype=torch.float32)
    x = torch.randn(4, 3, 64, 64, device="cuda")
    out = eng(x)
    with torch.no_grad():
        want = ref(_cl(x))
    assert torch.allclose(out, want, atol=2e-2, rtol=1e-2), \
        (out - want).abs().max().item()


def test_packed_inference_hipgraph_replay():
    from bdbnn_amd.models import imagenet as im
    from bdbnn_amd.engine import PackedInference
    torch.manual_seed(13)
    model = im.resnet18(False)
    eng = PackedInference(model, dtype=torch.float32)
    x = torch.randn(2, 3, 64, 64, device="cuda")
    eager = eng(x).clone()
    eng.capture((2, 3, 64, 64))
    replayed = eng(x).clone()
    assert torch.allclose(eager, replayed, atol=1e-3, rtol=1e-3)
    # replay twice with different inputs gives different outputs
    x2 = torch.randn(2, 3, 64, 64, device="cuda")
    out2 = eng(x2).clone()
    assert not torch.allclose(replayed, out2)


# ---------------- packed-bit backward path ----------------

def test_sign_mask_pack_and_decode():
    torch.manual_seed(14)
    x = torch.randn(2, 64, 6, 6, device="cuda") * 2
    nat = _nat()
    sp, mp = nat.sign_mask_pack_nhwc(_cl(x))
    sp2 = nat.sign_pack_nhwc(_cl(x))
    assert torch.equal(sp, sp2)
    xb = nat.decode_packed(sp, 64, False)
    assert torch.equal(xb, _cl(binsign(x)))
    g = torch.randn(2, 64, 6, 6, device="cuda")
    dx = nat.mask_mul_packed(_cl(g), mp, 64, False)
    ref = _cl(g * (x.abs() <= 1).float())
    assert torch.equal(dx, ref)


def test_weight_decode_roundtrip():
    torch.manual_seed(15)
    w = torch.randn(32, 48, 3, 3, device="cuda")  # tail-word C
    nat = _nat()
    wp, alpha, stab = nat.weight_pack(w)
    wb = nat.weight_decode(wp, alpha, 48, False)
    ref = weight_scale(w) * binsign(w)
    assert torch.allclose(wb, ref, atol=1e-6)


# ---------------- fused maxpool ----------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_maxpool_matches_torch(dtype):
    from bdbnn_amd.ops.pool import FusedMaxPool2d
    torch.manual_seed(16)
    mp = FusedMaxPool2d(3, 2, 1)
    x = _cl(torch.randn(4, 64, 23, 23, device="cuda", dtype=dtype))
    x = x.requires_grad_(True)
    out = mp(x)
    x2 = x.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.max_pool2d(x2, 3, 2, 1)
    assert torch.equal(out, _cl(ref))
    g = torch.randn_like(out)
    out.backward(g)
    ref.backward(_cl(g))
    # ties in bf16 can route gradient to a different (equally maximal)
    # element; compare per-window sums instead for bf16
    if dtype == torch.float32:
        assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    else:
        assert torch.allclose(x.grad.float().sum(), x2.grad.float().sum(),
                              rtol=1e-2)


def test_conv_epilogue_stats_match_output_sums():
    torch.manual_seed(17)
    nat = _nat()
    x = torch.randn(3, 64, 14, 14, device="cuda")
    w = torch.randn(96, 64, 3, 3, device="cuda")
    xp = nat.sign_pack_nhwc(_cl(x))
    wp, alpha, stab = nat.weight_pack(w)
    out, s1, s2 = nat.xnor_conv_fwd(xp, wp, alpha, stab, 64, 1, 1, False,
                                    True)
    ref1 = out.sum(dim=(0, 2, 3))
    ref2 = (out * out).sum(dim=(0, 2, 3))
    # s1/s2 are [32][K] sliced partials (contention fix) — fold slices
    assert s1.shape == (32, 96) and s2.shape == (32, 96)
    assert torch.allclose(s1.sum(0), ref1, rtol=1e-4, atol=1e-2)
    assert torch.allclose(s2.sum(0), ref2, rtol=1e-4, atol=1e-1)


def test_block_with_fused_stats_matches_composition():
    """A whole BiBasicBlock forward+backward with conv-epilogue stats vs
    the CPU oracle."""
    from bdbnn_amd.models.resnet_common import BiBasicBlock
    torch.manual_seed(18)
    blk_cpu = BiBasicBlock(64, 64)
    blk_gpu = BiBasicBlock(64, 64)
    blk_gpu.load_state_dict(blk_cpu.state_dict())
    blk_gpu = blk_gpu.cuda().to(memory_format=torch.channels_last)
    x = torch.randn(2, 64, 10, 10)
    xg = _cl(x.cuda()).requires_grad_(True)
    xc = x.clone().requires_grad_(True)
    out_g = blk_gpu(xg)
    if isinstance(out_g, tuple):   # (out, pack hand-off) on the GPU path
        out_g = out_g[0]
    out_c = blk_cpu(xc)
    assert torch.allclose(out_g.cpu(), out_c, atol=5e-3, rtol=1e-3), \
        (out_g.cpu() - out_c).abs().max().item()
    g = torch.randn_like(out_c)
    out_g.backward(_cl(g.cuda()))
    out_c.backward(g)
    assert torch.allclose(xg.grad.cpu(), xc.grad, atol=5e-3, rtol=1e-2)
    for (n1, p1), (n2, p2) in zip(blk_gpu.named_parameters(),
                                  blk_cpu.named_parameters()):
        assert torch.allclose(p1.grad.cpu(), p2.grad, atol=5e-3,
                              rtol=1e-2), n1


# ---------------- fused classification losses ----------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_logit_kd(dtype):
    from bdbnn_amd.ops.losses import fused_logit_kd
    torch.manual_seed(19)
    s = (torch.randn(16, 1000, device="cuda", dtype=dtype) * 3
         ).requires_grad_(True)
    t = torch.randn(16, 1000, device="cuda", dtype=dtype) * 3
    loss = fused_logit_kd(s, t)
    s2 = s.detach().float().clone().requires_grad_(True)
    ref = -(torch.softmax(t.float(), 1)
            * torch.log_softmax(s2, 1)).sum(1).mean()
    atol = 1e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(loss.float(), ref, atol=atol, rtol=1e-3)
    loss.backward()
    ref.backward()
    assert torch.allclose(s.grad.float(), s2.grad,
                          atol=1e-5 if dtype == torch.float32 else 1e-3,
                          rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_cross_entropy(dtype):
    from bdbnn_amd.ops.losses import FusedCrossEntropy
    torch.manual_seed(20)
    s = (torch.randn(32, 1000, device="cuda", dtype=dtype) * 2
         ).requires_grad_(True)
    y = torch.randint(0, 1000, (32,), device="cuda")
    loss = FusedCrossEntropy()(s, y)
    s2 = s.detach().float().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(s2, y)
    atol = 1e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(loss.float(), ref, atol=atol, rtol=1e-3)
    g = torch.tensor(1.7, device="cuda")
    loss.backward(g)
    ref.backward(g.float())
    assert torch.allclose(s.grad.float(), s2.grad,
                          atol=1e-5 if dtype == torch.float32 else 1e-3,
                          rtol=1e-2)


def test_fused_bn_act_train_bf16():
    from bdbnn_amd.ops.bn_act import fused_bn_act
    from bdbnn_amd.ops.activations import ChannelPReLU
    torch.manual_seed(21)
    C = 64
    bn = torch.nn.BatchNorm2d(C).cuda()
    bn2 = torch.nn.BatchNorm2d(C).cuda()
    bn2.load_state_dict(bn.state_dict())
    act = ChannelPReLU(C).cuda()
    x = _cl(torch.randn(8, C, 14, 14, device="cuda",
                        dtype=torch.bfloat16) * 2).requires_grad_(True)
    skip = _cl(torch.randn_like(x.detach())).requires_grad_(True)
    out = fused_bn_act(x, bn, act, skip=skip)
    x2 = x.detach().float().clone().requires_grad_(True)
    s2 = skip.detach().float().clone().requires_grad_(True)
    z = bn2(x2) + s2
    ref = torch.nn.functional.prelu(z, act.weight.detach())
    assert torch.allclose(out.float(), _cl(ref), atol=5e-2, rtol=1e-2)
    g = torch.randn_like(out)
    out.backward(g)
    ref.backward(_cl(g.float()))
    assert torch.allclose(x.grad.float(), x2.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(skip.grad.float(), s2.grad, atol=5e-2, rtol=5e-2)


def test_model_eval_gpu_matches_cpu():
    from bdbnn_amd.models import imagenet as im
    torch.manual_seed(22)
    m_cpu = im.resnet18(False)
    # populate running stats with a couple of train steps
    for _ in range(2):
        m_cpu(torch.randn(4, 3, 64, 64))
    m_gpu = im.resnet18(False)
    m_gpu.load_state_dict(m_cpu.state_dict())
    m_gpu = m_gpu.cuda().to(memory_format=torch.channels_last).eval()
    m_cpu.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        out_g = m_gpu(_cl(x.cuda()))
        out_c = m_cpu(x)
    assert torch.allclose(out_g.cpu(), out_c, atol=2e-2, rtol=1e-2), \
        (out_g.cpu() - out_c).abs().max().item()


# ---------------- experimental MFMA dgrad (round-2 seed) ----------------

import os as _os


def test_experimental_conv_dgrad_matches_reference():
    # validated on MI355X (first-shot numerics pass); default-off in the
    # training path until round-2 perf tuning
    torch.manual_seed(23)
    nat = _nat()
    for (N, C, H, K) in [(2, 64, 8, 64), (1, 128, 16, 32)]:
        g = torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16)
        g = _cl(g)
        w = torch.randn(K, C, 3, 3, device="cuda")
        wp, alpha, stab = nat.weight_pack(w)
        if (N * H * H) % 128 or C % 64 or K % 16:
            continue
        dx = nat.conv_dgrad(g, wp, alpha, C)
        wb = (weight_scale(w) * binsign(w)).to(torch.bfloat16)
        ref = torch.nn.functional.conv_transpose2d(
            g.float(), wb.float(), None, stride=1, padding=1)
        assert dx.shape == ref.shape
        assert torch.allclose(dx.float(), _cl(ref), atol=0.5, rtol=2e-2), \
            (dx.float() - _cl(ref)).abs().max().item()


def test_dgrad_v1_v2_cross_check():
    """The two independent MFMA dgrad implementations (v1 rocWMMA
    in-kernel decode, v2 halo implicit GEMM) must agree: v2's fused
    mask == mask_mul(v1)."""
    torch.manual_seed(25)
    nat = _nat()
    N, C, H, K = 2, 64, 56, 64
    g = _cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(K, C, 3, 3, device="cuda")
    x = torch.randn(N, C, H, H, device="cuda")
    wp, alpha, _ = nat.weight_pack(w)
    _, mp = nat.sign_mask_pack_nhwc(_cl(x))
    wd = nat.dgrad_weight_decode(wp, alpha, C)
    dx2 = nat.conv_dgrad2(g, wd, mp, C)
    dx1 = nat.mask_mul_packed(nat.conv_dgrad(g, wp, alpha, C), mp, C, True)
    assert torch.allclose(dx2.float(), dx1.float(), atol=5e-2,
                          rtol=2e-2), (dx2.float() -
                                       dx1.float()).abs().max().item()


def test_experimental_conv_wgrad_matches_reference():
    # validated on MI355X (first-shot numerics pass); default-off in the
    # training path until round-2 perf tuning
    torch.manual_seed(24)
    nat = _nat()
    for (N, C, H, K) in [(2, 32, 8, 32), (1, 64, 16, 64)]:
        x = torch.randn(N, C, H, H, device="cuda")
        g = _cl(torch.randn(N, K, H, H, device="cuda",
                            dtype=torch.bfloat16))
        xp = nat.sign_pack_nhwc(_cl(x))
        dw = nat.conv_wgrad(g, xp, C)
        xb = binsign(x).to(torch.bfloat16)
        ref = torch.ops.aten.convolution_backward(
            g.float(), xb.float(),
            torch.empty(K, C, 3, 3, device="cuda"), None,
            [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
        assert dw.shape == ref.shape
        assert torch.allclose(dw, ref, atol=0.5, rtol=2e-2), \
            (dw - ref).abs().max().item()


def test_vgg_small_gpu_step():
    from bdbnn_amd.models.cifar10 import vgg_small
    from bdbnn_amd.ops.optim import FusedAdam
    torch.manual_seed(25)
    m = vgg_small().cuda().to(memory_format=torch.channels_last)
    opt = FusedAdam(m.parameters(), lr=1e-3)
    x = _cl(torch.randn(8, 3, 32, 32, device="cuda"))
    y = torch.randint(0, 10, (8,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(m(x), y)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)


# ---------------- MFMA dgrad v2 (hot path) ----------------

def _dgrad2_ref(g, w, mask):
    """fp32 oracle: conv_transpose with alpha*sign(w) + clip-STE mask."""
    wb = (weight_scale(w) * binsign(w)).to(torch.bfloat16)
    ref = torch.nn.functional.conv_transpose2d(
        g.float(), wb.float(), None, stride=1, padding=1)
    return ref * mask.float()


@pytest.mark.parametrize("N,C,H,K", [
    (2, 64, 56, 64),     # Wp=64 class (stage 1)
    (2, 128, 28, 128),   # Wp=32
    (2, 256, 14, 256),   # Wp=16
    (4, 512, 7, 512),    # Wp=8 (multi-image blocks)
    (1, 64, 9, 64),      # Wp=16 with W<Wp (dummy columns + partial band)
    (3, 64, 30, 64),     # Wp=32, H%Rb != 0 (dummy band rows), odd N
])
def test_conv_dgrad2_matches_reference(N, C, H, K):
    torch.manual_seed(31)
    nat = _nat()
    g = _cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(K, C, 3, 3, device="cuda")
    x = torch.randn(N, C, H, H, device="cuda")
    wp, alpha, _ = nat.weight_pack(w)
    _, mp = nat.sign_mask_pack_nhwc(_cl(x))
    assert nat.dgrad2_supported(H, H, C, K)
    wd = nat.dgrad_weight_decode(wp, alpha, C)
    dx = nat.conv_dgrad2(g, wd, mp, C)
    mask = (x.abs() <= 1).to(torch.float32)
    ref = _dgrad2_ref(g, w, mask)
    assert dx.shape == ref.shape
    err = (dx.float() - _cl(ref)).abs().max().item()
    assert torch.allclose(dx.float(), _cl(ref), atol=0.5, rtol=2e-2), err


@pytest.mark.parametrize("N,C,H,K", [(2, 64, 56, 64), (4, 512, 7, 512)])
def test_conv_dgrad2_acc_fuses_skip_grad(N, C, H, K):
    """The optional acc operand must equal dgrad2 + a separate add of
    the (unmasked) skip gradient."""
    torch.manual_seed(33)
    nat = _nat()
    g = _cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(K, C, 3, 3, device="cuda")
    x = torch.randn(N, C, H, H, device="cuda")
    skip_g = _cl(torch.randn(N, C, H, H, device="cuda",
                             dtype=torch.bfloat16))
    wp, alpha, _ = nat.weight_pack(w)
    _, mp = nat.sign_mask_pack_nhwc(_cl(x))
    wd = nat.dgrad_weight_decode(wp, alpha, C)
    base = nat.conv_dgrad2(g, wd, mp, C)
    fused = nat.conv_dgrad2(g, wd, mp, C, skip_g)
    want = (base.float() + skip_g.float()).to(torch.bfloat16)
    assert torch.allclose(fused.float(), want.float(), atol=2e-2,
                          rtol=1e-2), (fused.float() -
                                       want.float()).abs().max().item()


def test_bi_block_defer_skip_grad_matches_plain():
    """BiBasicBlock grads with the deferred-skip-grad path (default on)
    vs plain autograd accumulation, same weights/inputs."""
    from bdbnn_amd.models import resnet_common as rc
    torch.manual_seed(34)
    blk = rc.BiBasicBlock(64, 64).cuda().to(
        memory_format=torch.channels_last)
    x0 = torch.randn(2, 64, 14, 14, device="cuda")
    gout = torch.randn(2, 64, 14, 14, device="cuda")

    def run(defer):
        old = rc._FUSE_SKIP_GRAD
        rc._FUSE_SKIP_GRAD = defer
        try:
            blk.zero_grad(set_to_none=True)
            x = _cl(x0.clone()).requires_grad_(True)
            out = blk(x)
            if isinstance(out, tuple):
                out = out[0]
            out.backward(_cl(gout))
            return ([p.grad.clone() for p in blk.parameters()],
                    x.grad.clone())
        finally:
            rc._FUSE_SKIP_GRAD = old

    pg_d, xg_d = run(True)
    pg_p, xg_p = run(False)
    assert torch.allclose(xg_d, xg_p, atol=1e-4, rtol=1e-3), \
        (xg_d - xg_p).abs().max().item()
    for a, b in zip(pg_d, pg_p):
        assert torch.allclose(a, b, atol=1e-4, rtol=1e-3)


def test_bi_block_defer_skip_grad_bf16_dgrad2_path():
    """Same as above but bf16/autocast on a dgrad2-supported shape, so
    the deferred skip grad rides the fused conv_dgrad2 acc operand."""
    from bdbnn_amd.models import resnet_common as rc
    torch.manual_seed(35)
    blk = rc.BiBasicBlock(64, 64).cuda().to(
        memory_format=torch.channels_last)
    x0 = torch.randn(2, 64, 56, 56, device="cuda", dtype=torch.bfloat16)
    gout = torch.randn(2, 64, 56, 56, device="cuda",
                       dtype=torch.bfloat16)

    def run(defer):
        old = rc._FUSE_SKIP_GRAD
        rc._FUSE_SKIP_GRAD = defer
        try:
            blk.zero_grad(set_to_none=True)
            x = _cl(x0.clone()).requires_grad_(True)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = blk(x)
            if isinstance(out, tuple):
                out = out[0]
            out.backward(_cl(gout))
            return ([p.grad.clone() for p in blk.parameters()],
                    x.grad.clone())
        finally:
            rc._FUSE_SKIP_GRAD = old

    pg_d, xg_d = run(True)
    pg_p, xg_p = run(False)
    # only rounding differs: fused adds in f32 then rounds once; plain
    # adds two already-rounded bf16 tensors.  The input grad (the direct
    # consumer of the fused add) must agree elementwise; param grads sit
    # behind large bf16 reductions of that rounded tensor where
    # elementwise cancellation amplifies the rounding delta, so they are
    # compared in relative norm (a routing bug — lost/duplicated dskip —
    # would show up as an O(1) relative error; the fp32 variant above
    # checks exact routing at 1e-4).
    assert torch.allclose(xg_d.float(), xg_p.float(), atol=3e-2,
                          rtol=2e-2), (xg_d.float() -
                                       xg_p.float()).abs().max().item()
    for a, b in zip(pg_d, pg_p):
        rel = (a.float() - b.float()).norm() / (b.float().norm() + 1e-12)
        assert rel.item() < 0.05, rel.item()


def test_dgrad_weight_decode_values():
    torch.manual_seed(32)
    nat = _nat()
    K, C = 64, 64
    w = torch.randn(K, C, 3, 3, device="cuda")
    wp, alpha, _ = nat.weight_pack(w)
    wd = nat.dgrad_weight_decode(wp, alpha, C)   # [9][C][K]
    want = (weight_scale(w) * binsign(w)).to(torch.bfloat16)  # [K][C][3][3]
    for t in range(9):
        dy, dx_ = t // 3, t % 3
        ref = want[:, :, 2 - dy, 2 - dx_].transpose(0, 1)  # [C][K] mirrored
        assert torch.equal(wd[t], ref)


def test_packed_backward_uses_mfma_dgrad():
    """Full BinaryConvFunction backward: MFMA-on vs MIOpen fallback agree."""
    import bdbnn_amd.ops.binary_conv as bc
    torch.manual_seed(33)
    x = _cl(torch.randn(2, 64, 14, 14, device="cuda",
                        dtype=torch.bfloat16))
    w = torch.randn(64, 64, 3, 3, device="cuda")

    def run(flag):
        old = bc._MFMA_BWD
        bc._MFMA_BWD = flag
        try:
            xl = x.clone().requires_grad_(True)
            wl = w.clone().requires_grad_(True)
            out, _, _ = BinaryConvFunction.apply(xl, wl, 1, 1, "ste", None,
                                                 None, False)
            out.float().pow(2).sum().backward()
            return xl.grad.float(), wl.grad.float()
        finally:
            bc._MFMA_BWD = old
    dx2, dw2 = run(True)
    dx1, dw1 = run(False)
    assert torch.allclose(dx2, dx1, atol=1e-2, rtol=1e-2), \
        (dx2 - dx1).abs().max().item()
    assert torch.allclose(dw2, dw1, atol=1e-2, rtol=1e-2)


# ---------------- MFMA wgrad v2 (hot path) ----------------

@pytest.mark.parametrize("N,C,H,K", [
    (2, 64, 56, 64),
    (2, 128, 28, 128),
    (2, 256, 14, 256),
    (4, 512, 7, 512),
    (1, 64, 9, 64),      # W < Wp (dummy columns)
    (3, 64, 30, 64),     # H % rc != 0, odd N
])
def test_conv_wgrad2_matches_reference(N, C, H, K):
    torch.manual_seed(41)
    nat = _nat()
    x = torch.randn(N, C, H, H, device="cuda")
    g = _cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
    xp = nat.sign_pack_nhwc(_cl(x))
    xcp = nat.repack_cplane(xp, C, H)
    dwT = nat.conv_wgrad2(g, xcp, C).sum(0)    # [nslab][9][C][K] fp32
    xb = binsign(x).to(torch.bfloat16)
    ref = torch.ops.aten.convolution_backward(
        g.float(), _cl(xb.float()),
        torch.empty(K, C, 3, 3, device="cuda"), None,
        [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
        [False, True, False])[1]               # [K][C][3][3]
    got = dwT.permute(2, 1, 0).reshape(K, C, 3, 3)
    err = (got - ref).abs().max().item()
    # fp32 atomic accumulation of bf16 products vs MIOpen's order: small
    # relative slack on sums over N*H*W terms
    assert torch.allclose(got, ref, atol=2.0, rtol=2e-2), err


def test_repack_cplane_bits():
    torch.manual_seed(42)
    nat = _nat()
    N, C, H, W = 2, 64, 5, 9
    x = torch.randn(N, C, H, W, device="cuda")
    xp = nat.sign_pack_nhwc(_cl(x))
    xcp = nat.repack_cplane(xp, C, W).cpu().numpy().astype("uint64")
    xs = x.permute(0, 2, 3, 1).cpu()
    for c in range(0, C, 17):
        for n in range(N):
            for y in range(H):
                row = int(xcp[c][n * H + y])
                for xx in range(64):
                    bit = (row >> xx) & 1
                    want = (1 if xs[n, y, xx, c].item() >= 0 else 0) \
                        if xx < W else 0
                    assert bit == want, (c, n, y, xx)


def test_wgrad_finish_transpose_and_mask():
    torch.manual_seed(43)
    nat = _nat()
    K, C = 64, 64
    dwT = torch.randn(3, 9, C, K, device="cuda")   # 3 m-split slabs
    w = (torch.randn(K, C, 3, 3, device="cuda") * 1.2)
    dw = nat.wgrad_finish(dwT, w)
    ref = (dwT.sum(0).permute(2, 1, 0).reshape(K, C, 3, 3)
           * (w.abs() <= 1).float())
    assert torch.allclose(dw, ref, atol=1e-5)


def test_full_packed_backward_mfma_vs_miopen():
    """Both dx and dw of the default path vs the BDBNN_MFMA_BWD=0 path."""
    import bdbnn_amd.ops.binary_conv as bc
    torch.manual_seed(44)
    for (N, C, H, K) in [(2, 64, 28, 64), (2, 64, 7, 128)]:
        x = _cl(torch.randn(N, C, H, H, device="cuda",
                            dtype=torch.bfloat16))
        w = torch.randn(K, C, 3, 3, device="cuda")

        def run(flag):
            old = bc._MFMA_BWD
            bc._MFMA_BWD = flag
            try:
                xl = x.clone().requires_grad_(True)
                wl = w.clone().requires_grad_(True)
                out, _, _ = BinaryConvFunction.apply(
                    xl, wl, 1, 1, "ste", None, None, False)
                (out.float() * torch.randn_like(out.float())).sum().backward()
                return xl.grad.float(), wl.grad.float()
            finally:
                bc._MFMA_BWD = old
        torch.manual_seed(45)
        dx2, dw2 = run(True)
        torch.manual_seed(45)
        dx1, dw1 = run(False)
        assert torch.allclose(dx2, dx1, atol=5e-2, rtol=2e-2), \
            (dx2 - dx1).abs().max().item()
        assert torch.allclose(dw2, dw1, atol=5e-1, rtol=2e-2), \
            (dw2 - dw1).abs().max().item()


# ---------------- BN-epilogue pack fusion ----------------

def test_bn_epilogue_pack_matches_pack_kernel():
    """bn_act_fwd_train(want_pack=True) bitplanes == sign_mask_pack of
    the written output, bit-exact, both dtypes and all act kinds."""
    nat = _nat()
    torch.manual_seed(45)
    for dtype in (torch.bfloat16, torch.float32):
        for act_kind in (0, 1, 2):
            x = _cl(torch.randn(4, 64, 14, 14, device="cuda",
                                dtype=dtype)) * 1.5
            gamma = torch.randn(64, device="cuda").abs() + 0.5
            beta = torch.randn(64, device="cuda") * 0.1
            a = torch.rand(64, device="cuda") * 0.3
            rm = torch.zeros(64, device="cuda")
            rv = torch.ones(64, device="cuda")
            res = nat.bn_act_fwd_train(x, None, gamma, beta,
                                       a if act_kind == 1 else None,
                                       rm, rv, 0.1, 1e-5, act_kind,
                                       None, None, True)
            out, xpk, mpk = res[0], res[4], res[5]
            xp_ref, mp_ref = nat.sign_mask_pack_nhwc(out)
            assert torch.equal(xpk.flatten(), xp_ref.flatten()), \
                (dtype, act_kind)
            assert torch.equal(mpk.flatten(), mp_ref.flatten()), \
                (dtype, act_kind)


def test_resnet_block_chain_with_pack_fusion():
    """Two chained BiBasicBlocks on GPU (pack hand-off across blocks)
    against the CPU oracle: same loss, close grads."""
    from bdbnn_amd.models.resnet_common import BiBasicBlock
    torch.manual_seed(46)
    chain_cpu = torch.nn.Sequential(BiBasicBlock(64, 64),
                                    BiBasicBlock(64, 64))
    chain_gpu = torch.nn.Sequential(BiBasicBlock(64, 64),
                                    BiBasicBlock(64, 64))
    chain_gpu.load_state_dict(chain_cpu.state_dict())
    chain_gpu = chain_gpu.cuda().to(memory_format=torch.channels_last)
    x = torch.randn(4, 64, 14, 14)
    xg = _cl(x.cuda()).requires_grad_(True)
    xc = x.clone().requires_grad_(True)
    out_g = chain_gpu(xg)
    if isinstance(out_g, tuple):
        out_g = out_g[0]
    out_c = chain_cpu(xc)
    lg = out_g.float().pow(2).mean()
    lc = out_c.pow(2).mean()
    assert abs(lg.item() - lc.item()) < 5e-3, (lg.item(), lc.item())
    lg.backward()
    lc.backward()
    gw_g = chain_gpu[0].conv1.weight.grad.cpu()
    gw_c = chain_cpu[0].conv1.weight.grad
    assert torch.allclose(gw_g, gw_c, atol=5e-3, rtol=5e-2), \
        (gw_g - gw_c).abs().max().item()


# ---------------- MFMA stem conv (7x7/2, 3->64) ----------------

def test_stem_conv_fwd_matches_reference():
    nat = _nat()
    torch.manual_seed(47)
    for (N, H) in [(2, 224), (3, 56)]:
        x = _cl(torch.randn(N, 3, H, H, device="cuda",
                            dtype=torch.bfloat16))
        w = torch.randn(64, 3, 7, 7, device="cuda") * 0.1
        out, x4 = nat.stem_conv_fwd(x, w)
        assert x4.shape == (N, H, H, 4)
        ref = torch.nn.functional.conv2d(
            x.float(), w, None, stride=2, padding=3)
        assert out.shape == ref.shape
        err = (out.float() - _cl(ref)).abs().max().item()
        assert torch.allclose(out.float(), _cl(ref), atol=0.1,
                              rtol=2e-2), (H, err)


def test_stem_conv_wrw_matches_reference():
    nat = _nat()
    torch.manual_seed(48)
    for (N, H) in [(2, 224), (3, 56)]:
        x = _cl(torch.randn(N, 3, H, H, device="cuda",
                            dtype=torch.bfloat16))
        w = torch.randn(64, 3, 7, 7, device="cuda") * 0.1
        _, x4 = nat.stem_conv_fwd(x, w)
        g = _cl(torch.randn(N, 64, H // 2, H // 2, device="cuda",
                            dtype=torch.bfloat16))
        dw = nat.stem_conv_wrw(x4, g)
        ref = torch.ops.aten.convolution_backward(
            g.float(), x.float(), torch.empty(64, 3, 7, 7, device="cuda"),
            None, [2, 2], [3, 3], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
        err = (dw - ref).abs().max().item()
        rel = err / ref.abs().max().item()
        assert rel < 2e-2, (H, err, rel)


def test_stem_module_autograd():
    """StemConv7x7 module: fast path out + dw vs the stock conv path."""
    from bdbnn_amd.ops.stem_conv import StemConv7x7
    torch.manual_seed(49)
    m = StemConv7x7(3, 64, 7, 2, 3, bias=False).cuda()
    m = m.to(memory_format=torch.channels_last)
    x = _cl(torch.randn(4, 3, 64, 64, device="cuda"))
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = m(x)
    out.float().pow(2).mean().backward()
    dw_fast = m.weight.grad.clone()
    m.weight.grad = None
    ref_out = torch.nn.functional.conv2d(
        x.to(torch.bfloat16), m.weight.to(torch.bfloat16), None,
        stride=2, padding=3)
    ref_out.float().pow(2).mean().backward()
    dw_ref = m.weight.grad
    assert torch.allclose(out.float(), ref_out.float(), atol=0.1,
                          rtol=2e-2)
    rel = (dw_fast - dw_ref).abs().max() / dw_ref.abs().max()
    assert rel.item() < 3e-2, rel.item()

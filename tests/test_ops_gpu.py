"""GPU kernel numerics: every HIP kernel vs the plain-PyTorch fp32 oracle
(the same implementations that run on the CPU path).  All tests @gpu."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def to_gpu(*ts, dtype=None):
    out = []
    for t in ts:
        t = t.cuda()
        if dtype is not None:
            t = t.to(dtype)
        out.append(t)
    return out[0] if len(out) == 1 else out


# ---------------------------------------------------------------------------
# l2norm
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(32, 128), (7, 128), (256, 128), (4, 64)])
def test_l2norm_fwd_bwd(dtype, shape):
    from moco_amd import _C
    torch.manual_seed(0)
    x = torch.randn(*shape)
    xg = x.cuda().to(dtype)
    y, inv = _C.l2norm_rows_fwd(xg)
    ref = torch.nn.functional.normalize(x.float(), dim=1)
    atol = 1e-5 if dtype == torch.float32 else 2e-2
    assert torch.allclose(y.float().cpu(), ref, atol=atol)
    # bf16 inputs round before the norm: ~0.4% relative slack
    inv_rtol = 1e-4 if dtype == torch.float32 else 2e-2
    assert torch.allclose(inv.cpu(), 1.0 / x.float().norm(dim=1),
                          rtol=inv_rtol)

    g = torch.randn(*shape)
    gg = g.cuda().to(dtype)
    gx = _C.l2norm_rows_bwd(gg, y, inv)
    x2 = x.clone().requires_grad_(True)
    torch.nn.functional.normalize(x2, dim=1).backward(g)
    assert torch.allclose(gx.float().cpu(), x2.grad, atol=5e-2
                          if dtype == torch.bfloat16 else 1e-5)


# ---------------------------------------------------------------------------
# contrastive logits (MFMA layout check: random asymmetric data)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("N,C,K", [(32, 128, 1024), (8, 128, 64),
                                   (256, 128, 65536), (7, 128, 256),
                                   (16, 64, 512), (33, 256, 4096)])
def test_contrastive_fwd(N, C, K):
    from moco_amd import _C
    torch.manual_seed(1)
    invT = 1.0 / 0.2
    # unit-norm rows, like real usage
    q = torch.nn.functional.normalize(torch.randn(N, C), dim=1)
    k = torch.nn.functional.normalize(torch.randn(N, C), dim=1)
    queue_kc = torch.nn.functional.normalize(torch.randn(K, C), dim=1)
    qg, kg, Qg = to_gpu(q, k, queue_kc, dtype=torch.bfloat16)
    out = _C.contrastive_fwd(qg, kg, Qg, invT)
    assert out.shape == (N, 1 + K) and out.dtype == torch.float32
    l_pos = (q * k).sum(1, keepdim=True)
    l_neg = q @ queue_kc.t()
    ref = torch.cat([l_pos, l_neg], 1) * invT
    # bf16 inputs, fp32 accumulate: |logit| <= 1/T = 5
    assert torch.allclose(out.cpu(), ref, atol=8e-2), \
        (out.cpu() - ref).abs().max()


def test_contrastive_identity_layout():
    """A = one-hot rows against asymmetric queue: catches any fragment
    row/col transpose (guide G9/rule 16)."""
    from moco_amd import _C
    N, C, K = 16, 128, 64
    q = torch.zeros(N, C)
    for n in range(N):
        q[n, n + 3] = 1.0  # one-hot at shifted positions
    queue_kc = torch.arange(K * C, dtype=torch.float32).view(K, C) / (K * C)
    k = torch.zeros(N, C)
    qg, kg, Qg = to_gpu(q, k, queue_kc, dtype=torch.bfloat16)
    out = _C.contrastive_fwd(qg, kg, Qg, 1.0)
    ref = q @ queue_kc.to(torch.bfloat16).float().t()
    assert torch.allclose(out[:, 1:].cpu(), ref, atol=1e-3), \
        (out[:, 1:].cpu() - ref).abs().max()


@pytest.mark.parametrize("N,C,K", [(32, 128, 1024), (8, 128, 64),
                                   (256, 128, 65536), (7, 128, 4096)])
def test_contrastive_bwd(N, C, K):
    from moco_amd import _C
    torch.manual_seed(2)
    invT = 1.0 / 0.2
    g = torch.randn(N, 1 + K) / math.sqrt(K)  # CE-grad-like magnitude
    k = torch.nn.functional.normalize(torch.randn(N, C), dim=1)
    queue_ck = torch.nn.functional.normalize(torch.randn(C, K), dim=0)
    gg = g.cuda()
    kg = k.cuda().to(torch.bfloat16)
    Qg = queue_ck.cuda()
    dq = _C.contrastive_bwd(gg, kg, Qg, invT)
    ref = (g[:, :1] * k + g[:, 1:] @ queue_ck.t()) * invT
    assert torch.allclose(dq.cpu(), ref, atol=5e-2,
                          rtol=1e-2), (dq.cpu() - ref).abs().max()


# ---------------------------------------------------------------------------
# softmax CE (label 0)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("N,M", [(32, 1025), (8, 65537), (256, 65537),
                                 (7, 131073), (5, 13)])
def test_softmax_ce0(N, M):
    from moco_amd import _C
    torch.manual_seed(3)
    logits = (torch.randn(N, M) * 3).cuda()
    loss_rows, row_max, row_lse = _C.softmax_ce0_fwd(logits)
    ref_loss = torch.nn.functional.cross_entropy(
        logits.cpu(), torch.zeros(N, dtype=torch.long), reduction="none")
    assert torch.allclose(loss_rows.cpu(), ref_loss, atol=1e-4, rtol=1e-5)

    scale = 1.0 / N
    d = _C.softmax_ce0_bwd(logits, row_max, row_lse, scale)
    l2 = logits.cpu().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(
        l2, torch.zeros(N, dtype=torch.long)).backward()
    assert torch.allclose(d.cpu(), l2.grad, atol=1e-6), \
        (d.cpu() - l2.grad).abs().max()


# ---------------------------------------------------------------------------
# queue enqueue
# ---------------------------------------------------------------------------

def test_queue_enqueue_gpu():
    from moco_amd import _C
    C, K, B = 128, 256, 32
    torch.manual_seed(4)
    queue = torch.randn(C, K).cuda()
    queue_kc = queue.t().contiguous().to(torch.bfloat16)
    keys = torch.nn.functional.normalize(torch.randn(B, C), dim=1) \
        .cuda().to(torch.bfloat16)
    ref = queue.clone()
    _C.queue_enqueue(queue, queue_kc, keys, 64)
    ref[:, 64:64 + B] = keys.float().t()
    assert torch.allclose(queue, ref)
    assert torch.allclose(queue_kc[64:64 + B].float(), keys.float())
    # untouched rows of queue_kc intact
    assert torch.allclose(queue_kc[:64].float(),
                          queue[:, :64].t().to(torch.bfloat16).float())


# ---------------------------------------------------------------------------
# multi-tensor EMA / SGD
# ---------------------------------------------------------------------------

def test_multi_tensor_ema_gpu():
    from moco_amd.ops.multi_tensor import MultiTensorEMA
    torch.manual_seed(5)
    dst = [torch.randn(s).cuda() for s in [(7,), (130000,), (64, 64)]]
    src = [torch.randn_like(t) for t in dst]
    expect = [0.999 * d + 0.001 * s for d, s in zip(dst, src)]
    MultiTensorEMA(dst, src).step(0.999)
    torch.cuda.synchronize()
    for d, e in zip(dst, expect):
        assert torch.allclose(d, e, atol=1e-6)


def test_fused_sgd_gpu_matches_torch():
    from moco_amd.ops.multi_tensor import FusedSGD
    torch.manual_seed(6)
    shapes = [(64, 3, 7, 7), (1000,), (130001,)]
    p1 = [torch.randn(s).cuda().requires_grad_(True) for s in shapes]
    p2 = [p.detach().clone().requires_grad_(True) for p in p1]
    o1 = FusedSGD(p1, lr=0.1, momentum=0.9, weight_decay=1e-4)
    o2 = torch.optim.SGD(p2, lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(4):
        for a, b in zip(p1, p2):
            g = torch.randn_like(a)
            a.grad = g.clone()
            b.grad = g.clone()
        o1.step()
        o2.step()
    torch.cuda.synchronize()
    for a, b in zip(p1, p2):
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()


# ---------------------------------------------------------------------------
# augmentation pipeline vs torch oracle
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("aug_plus", [False, True])
def test_two_crop_augment_vs_oracle(aug_plus):
    from moco_amd import _C
    from moco_amd.ops.augment import _augment_torch, sample_two_crop_params
    torch.manual_seed(7)
    B, H, W, S = 4, 96, 96, 64
    x = torch.rand(B, 3, H, W)
    params = sample_two_crop_params(B, H, W, S, aug_plus)
    out = _C.two_crop_augment(x.cuda(), params.cuda(), S, False, False)
    ref = _augment_torch(x, params, S)
    diff = (out.cpu() - ref).abs()
    assert diff.max() < 5e-2, diff.max()
    assert (diff > 1e-2).float().mean() < 2e-3


def test_two_crop_augment_bf16_channels_last():
    from moco_amd import _C
    from moco_amd.ops.augment import _augment_torch, sample_two_crop_params
    torch.manual_seed(8)
    B, H, W, S = 2, 64, 64, 32
    x = torch.rand(B, 3, H, W)
    params = sample_two_crop_params(B, H, W, S, True)
    out = _C.two_crop_augment(x.cuda(), params.cuda(), S, True, True)
    assert out.dtype == torch.bfloat16
    assert out.is_contiguous(memory_format=torch.channels_last)
    ref = _augment_torch(x, params, S)
    assert (out.float().cpu() - ref).abs().max() < 8e-2


# ---------------------------------------------------------------------------
# end-to-end on GPU
# ---------------------------------------------------------------------------

def test_moco_train_step_gpu():
    import moco_amd
    from moco_amd.engine import MoCoStepper
    from moco_amd.models.resnet import resnet18
    from moco_amd.ops import FusedSGD

    torch.manual_seed(0)
    device = torch.device("cuda", 0)
    model = moco_amd.MoCo(
        lambda num_classes: resnet18(num_classes=num_classes),
        dim=128, K=1024, m=0.999, T=0.2, mlp=True)
    model = model.to(device).to(memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.03, momentum=0.9,
                   weight_decay=1e-4)
    stepper = MoCoStepper(model, opt, device, aug_plus=True, out_size=64)
    for i in range(3):
        raw = torch.rand(16, 3, 96, 96, device=device)
        loss, logits, labels = stepper.step_from_raw(raw)
    torch.cuda.synchronize()
    lv = float(loss.detach())
    assert lv == lv and 0 < lv < 100
    assert model.queue_ptr.item() == 48 % 1024


def test_moco_gpu_matches_cpu_reference_step():
    """One full MoCo forward on GPU (HIP kernels, bf16) vs the identical
    model on CPU (fp32 oracle): logits must agree to bf16 tolerance."""
    import moco_amd
    from moco_amd.models.resnet import resnet18

    torch.manual_seed(0)
    model_cpu = moco_amd.MoCo(
        lambda num_classes: resnet18(num_classes=num_classes),
        dim=128, K=256, m=0.999, T=0.2)
    model_gpu = moco_amd.MoCo(
        lambda num_classes: resnet18(num_classes=num_classes),
        dim=128, K=256, m=0.999, T=0.2)
    model_gpu.load_state_dict(model_cpu.state_dict())
    model_gpu = model_gpu.cuda()
    model_cpu.eval()
    model_gpu.eval()

    im_q = torch.randn(8, 3, 32, 32)
    im_k = torch.randn(8, 3, 32, 32)
    with torch.no_grad():
        lc, _ = model_cpu(im_q, im_k, step=0)
        lg, _ = model_gpu(im_q.cuda(), im_k.cuda(), step=0)
    # bf16 encoder + bf16 GEMM vs fp32: loose tolerance, but correlation
    # must be essentially perfect
    lcf = lc.flatten()
    lgf = lg.cpu().flatten()
    corr = torch.corrcoef(torch.stack([lcf, lgf]))[0, 1]
    assert corr > 0.999, corr
    assert (lcf - lgf).abs().max() < 0.5


# ---------------------------------------------------------------------------
# fused BatchNorm (+ReLU, +residual)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("C,HW,relu,res", [
    (64, 28, True, False), (128, 14, True, True), (256, 7, False, False),
    (512, 14, True, True), (2048, 7, True, False), (64, 112, True, False)])
def test_fused_bn_train_fwd_bwd(C, HW, relu, res):
    """Oracle = fp32 autograd on the SAME bf16-quantized inputs the kernel
    sees (bf16 input rounding shifts the ReLU mask for elements near zero,
    which is a property of bf16 activations, not of the kernel — quantizing
    the oracle's inputs removes that confound)."""
    import torch.nn as nn
    from moco_amd.ops.bn import bn_act
    torch.manual_seed(10)
    N = 8
    x = (torch.randn(N, C, HW, HW) * 2 + 0.3).to(torch.bfloat16).float()
    res_t = torch.randn(N, C, HW, HW).to(torch.bfloat16).float() \
        if res else None
    g = torch.randn(N, C, HW, HW).to(torch.bfloat16).float()

    bn_ref = nn.BatchNorm2d(C)
    bn_ref.weight.data.uniform_(0.5, 1.5)
    bn_ref.bias.data.uniform_(-0.3, 0.3)
    xr = x.clone().requires_grad_(True)
    rr = res_t.clone().requires_grad_(True) if res else None
    out = bn_ref(xr)
    if res:
        out = out + rr
    if relu:
        out = torch.relu(out)
    out.backward(g)

    # fused path: bf16 channels_last GPU
    bn_gpu = nn.BatchNorm2d(C).cuda()
    bn_gpu.load_state_dict(bn_ref.state_dict())
    bn_gpu.running_mean.zero_()
    bn_gpu.running_var.fill_(1.0)
    bn_gpu.num_batches_tracked.zero_()
    xg = x.cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    rg = None
    if res:
        rg = res_t.cuda().to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    yg = bn_act(xg, bn_gpu, relu=relu, residual=rg)
    yg.backward(g.cuda().to(torch.bfloat16)
                .contiguous(memory_format=torch.channels_last))

    def bulk_close(a, b, atol, frac=5e-5, what=""):
        d = (a - b).abs()
        assert float(d.mean()) < atol, (what, "mean", float(d.mean()))
        assert float((d > 20 * atol).float().mean()) < frac, \
            (what, "outliers", float((d > 20 * atol).float().mean()))

    bulk_close(yg.float().cpu(), out.detach(), 1e-2, what="y")
    assert torch.allclose(bn_gpu.running_mean.cpu(),
                          bn_ref.running_mean, atol=1e-2)
    assert torch.allclose(bn_gpu.running_var.cpu(),
                          bn_ref.running_var, atol=2e-2, rtol=1e-2)
    assert bn_gpu.num_batches_tracked.item() == 1
    M = float(x[:, 0].numel())
    # per-channel reductions of bf16-rounded terms: sqrt(M)-scaled slack
    tol = 2e-2 * max(1.0, float(bn_ref.weight.grad.abs().max())) \
        + 5e-3 * M ** 0.5
    assert (bn_gpu.weight.grad.cpu() - bn_ref.weight.grad).abs().max() < tol
    tol = 2e-2 * max(1.0, float(bn_ref.bias.grad.abs().max())) \
        + 5e-3 * M ** 0.5
    assert (bn_gpu.bias.grad.cpu() - bn_ref.bias.grad).abs().max() < tol
    bulk_close(xg.grad.float().cpu(), xr.grad, 1e-2, what="dx")
    if res:
        bulk_close(rg.grad.float().cpu(), rr.grad, 1e-2, what="dres")


def test_fused_bn_eval_matches_torch():
    import torch.nn as nn
    from moco_amd.ops.bn import bn_act
    torch.manual_seed(11)
    C = 128
    bn = nn.BatchNorm2d(C)
    bn.running_mean.normal_()
    bn.running_var.uniform_(0.5, 2.0)
    bn.weight.data.normal_(1.0, 0.2)
    bn.bias.data.normal_()
    bn.eval()
    x = torch.randn(4, C, 9, 9)
    ref = torch.relu(bn(x))
    bng = nn.BatchNorm2d(C).cuda()
    bng.load_state_dict(bn.state_dict())
    bng.eval()
    xg = x.cuda().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    y = bn_act(xg, bng, relu=True)
    assert (y.float().cpu() - ref).abs().max() < 6e-2
    assert bng.num_batches_tracked.item() == 0


# ---------------------------------------------------------------------------
# stem max-pool
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("N,C,H", [(4, 64, 112), (2, 64, 56), (3, 128, 27)])
def test_maxpool_fwd_bwd(N, C, H):
    from moco_amd.ops.pool import stem_maxpool
    torch.manual_seed(12)
    x = torch.randn(N, C, H, H).to(torch.bfloat16).float()
    xr = x.clone().requires_grad_(True)
    ref = torch.nn.functional.max_pool2d(xr, 3, stride=2, padding=1)
    g = torch.randn_like(ref)
    ref.backward(g)

    xg = x.cuda().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    y = stem_maxpool(xg)
    assert y.shape == ref.shape
    y.backward(g.cuda().to(torch.bfloat16)
               .contiguous(memory_format=torch.channels_last))
    assert (y.float().cpu() - ref.detach()).abs().max() < 1e-2
    # ties can route grad to a different (equal) tap; bf16 rounding can also
    # merge near-equal values: bound the bulk error
    d = (xg.grad.float().cpu() - xr.grad).abs()
    assert float(d.mean()) < 2e-3, float(d.mean())
    assert float((d > 0.2).float().mean()) < 1e-3


# ---------------------------------------------------------------------------
# 1x1-conv NT-GEMM (flag-gated path; numerics kept green for round 2)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("M,N,K", [(3136, 256, 64), (784, 64, 256),
                                   (100, 16, 32)])
def test_gemm_nt_bf16(M, N, K):
    from moco_amd import _C
    torch.manual_seed(13)
    A = (torch.randn(M, K) * 0.5).cuda().to(torch.bfloat16)
    Bt = (torch.randn(N, K) * 0.1).cuda().to(torch.bfloat16)
    C = _C.gemm_nt_bf16(A, Bt)
    ref = A.float() @ Bt.float().t()
    assert (C.float() - ref).abs().max() < 0.05 * max(1.0, K ** 0.5 * 0.05)


def test_conv1x1_function_matches_conv2d():
    import os
    os.environ["MOCO_AMD_CUSTOM_CONV1X1"] = "1"
    try:
        import torch.nn as nn
        from moco_amd.ops.conv1x1 import conv1x1
        torch.manual_seed(14)
        conv = nn.Conv2d(64, 128, 1, bias=False).cuda() \
            .to(memory_format=torch.channels_last)
        x = torch.randn(2, 64, 14, 14).cuda().to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last).requires_grad_(True)
        y = conv1x1(x, conv)
        ref = torch.nn.functional.conv2d(x.detach().float(),
                                         conv.weight.float())
        assert (y.float() - ref).abs().max() < 5e-2
        g = torch.randn_like(y)
        y.backward(g)
        xr = x.detach().float().requires_grad_(True)
        wr = conv.weight.detach().float().requires_grad_(True)
        torch.nn.functional.conv2d(xr, wr).backward(g.float())
        assert (x.grad.float() - xr.grad).abs().max() < 5e-2
        assert (conv.weight.grad.float() - wr.grad).abs().max() \
            < 0.1 * max(1.0, float(wr.grad.abs().max()))
    finally:
        os.environ.pop("MOCO_AMD_CUSTOM_CONV1X1", None)


def test_single_crop_augment_vs_oracle():
    from moco_amd.ops.augment import (_augment_torch, sample_two_crop_params,
                                      single_crop_augment)
    torch.manual_seed(15)
    B, H, W, S = 3, 64, 64, 32
    x = torch.rand(B, 3, H, W)
    params = sample_two_crop_params(B, H, W, S, False)[:B]
    out = single_crop_augment(x.cuda(), S, params.cuda())
    ref = _augment_torch(x, params, S)
    assert out.shape == (B, 3, S, S)
    assert (out.cpu() - ref).abs().max() < 5e-2


# ---------------------------------------------------------------------------
# fused GEMM + BN apply/stats (the conv+BN hot path) and remaining
# round-2 prototypes.  All in the default GPU suite.
# ---------------------------------------------------------------------------

def experimental(f):  # graduated round-2 kernels: no longer env-gated
    return f


@pytest.mark.parametrize("M,N,K", [(1536, 128, 64), (128, 64, 256)])
def test_gemm_nt_fused_apply_and_stats(M, N, K):
    from moco_amd import _C
    torch.manual_seed(16)
    A = (torch.randn(M, K) * 0.7 + 0.2).cuda().to(torch.bfloat16)
    Bt = (torch.randn(N, K) * 0.08).cuda().to(torch.bfloat16)
    scale = (torch.rand(K) * 1.5 + 0.25).cuda()
    shift = torch.randn(K).cuda() * 0.3
    C, ssum, ssq = _C.gemm_nt_fused(A, Bt, scale, shift, True, True)
    ssum, ssq = ssum.sum(0), ssq.sum(0)  # slot-striped partials
    # oracle: transform A (with bf16 re-rounding, as the kernel does), GEMM
    Af = A.float()
    At = torch.relu(Af * scale[None, :] + shift[None, :]) \
        .to(torch.bfloat16).float()
    ref = At @ Bt.float().t()
    assert (C.float() - ref).abs().max() < 0.06 * max(1.0, K ** 0.5 * 0.05)
    # stats are over the fp32 accumulator = ref columns
    assert torch.allclose(ssum.cpu(), ref.sum(0).cpu(), rtol=2e-3, atol=0.5)
    assert torch.allclose(ssq.cpu(), (ref ** 2).sum(0).cpu(), rtol=2e-3,
                          atol=1.0)


def test_gemm_nt_fused_no_options_matches_plain():
    from moco_amd import _C
    torch.manual_seed(17)
    A = torch.randn(256, 64).cuda().to(torch.bfloat16)
    Bt = torch.randn(128, 64).cuda().to(torch.bfloat16)
    (C,) = _C.gemm_nt_fused(A, Bt, None, None, False, False)
    ref = _C.gemm_nt_bf16(A, Bt)
    assert torch.equal(C, ref)


@pytest.mark.parametrize("N,Ci,H,Co,stride", [
    (8, 64, 16, 64, 1),     # M = 2048
    (8, 64, 16, 128, 1),
    (2, 128, 32, 64, 2),    # stride-2: Ho=16, M = 512
    (8, 256, 8, 192, 1),    # Co = 192 -> BNT 64 path
])
def test_conv3x3_fwd_vs_conv2d(N, Ci, H, Co, stride):
    from moco_amd import _C
    torch.manual_seed(18)
    x = (torch.randn(N, Ci, H, H) * 0.5).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Co, Ci, 3, 3) * 0.05).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    (out,) = _C.conv3x3_fwd(x, w, stride, None, None, False, False, None)
    ref = torch.nn.functional.conv2d(x.float(), w.float(), stride=stride,
                                     padding=1)
    assert out.shape == ref.shape
    tol = 0.05 * max(1.0, (9 * Ci) ** 0.5 * 0.025)
    assert (out.float() - ref).abs().max() < tol, \
        (out.float() - ref).abs().max()


def test_conv3x3_fused_apply_stats():
    """conv3x3 with the previous BN's apply folded in (neutral-page padding)
    + output stats."""
    from moco_amd import _C
    torch.manual_seed(19)
    N, Ci, H, Co = 8, 64, 16, 64
    x = (torch.randn(N, Ci, H, H)).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Co, Ci, 3, 3) * 0.05).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    scale = (torch.rand(Ci) + 0.5).cuda()
    shift = (torch.randn(Ci) * 0.2).cuda()
    neutral = torch.zeros(Ci + 8, dtype=torch.bfloat16, device="cuda")
    neutral[:Ci] = (-shift / scale).to(torch.bfloat16)
    out, ssum, ssq = _C.conv3x3_fwd(x, w, 1, scale, shift, True, True,
                                    neutral)
    # oracle: transform x (bf16 re-round), THEN zero-pad, then conv
    xt = torch.relu(x.float() * scale.view(1, -1, 1, 1)
                    + shift.view(1, -1, 1, 1)).to(torch.bfloat16).float()
    ref = torch.nn.functional.conv2d(xt, w.float(), padding=1)
    d = (out.float() - ref).abs()
    # neutral-page padding quantizes -shift/scale to bf16: border columns
    # carry a small extra error, bounded here
    assert d.max() < 0.15, d.max()
    assert d[:, :, 1:-1, 1:-1].max() < 0.06, d[:, :, 1:-1, 1:-1].max()
    assert torch.allclose(ssum.cpu(), ref.sum((0, 2, 3)).cpu(), rtol=5e-3,
                          atol=1.0)


def test_conv3x3_dgrad_s1_matches_autograd():
    from moco_amd.ops.conv3x3 import conv3x3_dgrad_s1
    torch.manual_seed(20)
    N, Ci, H, Co = 8, 64, 16, 64
    x = (torch.randn(N, Ci, H, H) * 0.5).cuda().float().requires_grad_(True)
    w = (torch.randn(Co, Ci, 3, 3) * 0.05).cuda().float()
    y = torch.nn.functional.conv2d(x, w, padding=1)
    g = torch.randn_like(y)
    y.backward(g)
    dx = conv3x3_dgrad_s1(
        g.to(torch.bfloat16).contiguous(memory_format=torch.channels_last),
        w.to(torch.bfloat16).contiguous(memory_format=torch.channels_last))
    tol = 0.05 * max(1.0, (9 * Co) ** 0.5 * 0.025)
    assert (dx.float() - x.grad).abs().max() < tol, \
        (dx.float() - x.grad).abs().max()


@pytest.mark.parametrize("N,Ci,H,Co,stride", [
    (8, 256, 16, 512, 2), (2, 64, 32, 128, 2), (8, 64, 16, 64, 1)])
def test_conv1x1_strided_vs_conv2d(N, Ci, H, Co, stride):
    from moco_amd import _C
    torch.manual_seed(21)
    x = (torch.randn(N, Ci, H, H) * 0.5).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Co, Ci, 1, 1) * 0.05).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    out = _C.conv1x1_strided(x, w, stride)
    ref = torch.nn.functional.conv2d(x.float(), w.float(), stride=stride)
    assert out.shape == ref.shape
    assert (out.float() - ref).abs().max() < 0.05 * max(1.0,
                                                        Ci ** 0.5 * 0.025)


def test_two_crop_augment_uint8_gpu():
    from moco_amd import _C
    from moco_amd.ops.augment import _augment_torch, sample_two_crop_params
    torch.manual_seed(22)
    B, H, W, S = 3, 64, 64, 32
    u8 = (torch.rand(B, 3, H, W) * 255).to(torch.uint8)
    params = sample_two_crop_params(B, H, W, S, True)
    out = _C.two_crop_augment(u8.cuda(), params.cuda(), S, False, False)
    ref = _augment_torch(u8.float() / 255.0, params, S)
    assert (out.cpu() - ref).abs().max() < 5e-2


@pytest.mark.parametrize("M,N,K", [(1536, 128, 128), (256, 64, 64)])
def test_gemm_nt_dgrad_epi(M, N, K):
    """dgrad GEMM whose epilogue applies the consumer ReLU mask and
    accumulates the BN-backward channel reductions."""
    from moco_amd import _C
    torch.manual_seed(23)
    A = (torch.randn(M, K) * 0.3).cuda().to(torch.bfloat16)
    Bt = (torch.randn(N, K) * 0.1).cuda().to(torch.bfloat16)
    y = torch.randn(M, N).cuda().to(torch.bfloat16)
    z = torch.randn(M, N).cuda().to(torch.bfloat16)
    mean = torch.randn(N).cuda() * 0.1
    rstd = (torch.rand(N) + 0.5).cuda()
    C, dbeta, dgamma = _C.gemm_nt_dgrad_epi(A, Bt, y, z, mean, rstd)
    dy = A.float() @ Bt.float().t()
    mask = (y.float() > 0).float()
    gp = (dy * mask).to(torch.bfloat16).float()
    tol = 0.05 * max(1.0, K ** 0.5 * 0.03)
    assert (C.float() - gp).abs().max() < tol
    # reductions use the pre-rounding fp32 g' (dy*mask), bf16-rounded stores
    ref_db = (dy * mask).sum(0)
    zhat = (z.float() - mean[None, :]) * rstd[None, :]
    ref_dg = (dy * mask * zhat).sum(0)
    assert torch.allclose(dbeta.cpu(), ref_db.cpu(), rtol=5e-3, atol=0.5)
    assert torch.allclose(dgamma.cpu(), ref_dg.cpu(), rtol=8e-3, atol=1.0)


# ---------------------------------------------------------------------------
# new round-2 kernel features: a_out / c_add epilogues, granular BN entries
# ---------------------------------------------------------------------------

def test_gemm_nt_fused_aout_materializes_transformed_a():
    from moco_amd import _C
    torch.manual_seed(24)
    M, N, K = 1536, 128, 64
    A = (torch.randn(M, K) * 0.7).cuda().to(torch.bfloat16)
    Bt = (torch.randn(N, K) * 0.08).cuda().to(torch.bfloat16)
    scale = (torch.rand(K) * 1.5 + 0.25).cuda()
    shift = (torch.randn(K) * 0.3).cuda()
    C, ssum, ssq, aout = _C.gemm_nt_fused(A, Bt, scale, shift, True, True,
                                          True, None)
    ssum, ssq = ssum.sum(0), ssq.sum(0)  # slot-striped partials
    ref_a = torch.relu(A.float() * scale[None, :] + shift[None, :]) \
        .to(torch.bfloat16)
    assert torch.equal(aout, ref_a)
    ref = ref_a.float() @ Bt.float().t()
    assert (C.float() - ref).abs().max() < 0.08


def test_gemm_nt_bf16_c_add_epilogue():
    from moco_amd import _C
    torch.manual_seed(25)
    M, N, K = 256, 64, 128
    A = (torch.randn(M, K) * 0.3).cuda().to(torch.bfloat16)
    Bt = (torch.randn(N, K) * 0.1).cuda().to(torch.bfloat16)
    add = torch.randn(M, N).cuda().to(torch.bfloat16)
    C = _C.gemm_nt_bf16(A, Bt, add)
    base = _C.gemm_nt_bf16(A, Bt)
    ref = (base.float() + add.float()).to(torch.bfloat16)
    assert torch.equal(C, ref)


def test_bn_finalize_matches_stats():
    from moco_amd import _C
    torch.manual_seed(26)
    R, C = 4096, 128
    x = torch.randn(R, C).cuda().to(torch.bfloat16)
    xf = x.float()
    ssum = xf.sum(0)
    ssq = (xf * xf).sum(0)
    rm = torch.zeros(C).cuda()
    rv = torch.ones(C).cuda()
    mean, rstd = _C.bn_finalize(ssum, ssq, R, rm, rv, 1e-5, 0.1)
    ref_mean = xf.mean(0)
    ref_var = xf.var(0, unbiased=False)
    assert torch.allclose(mean, ref_mean, atol=1e-3)
    assert torch.allclose(rstd, torch.rsqrt(ref_var + 1e-5), rtol=1e-3)
    assert torch.allclose(rm, 0.1 * ref_mean, atol=1e-3)
    assert torch.allclose(rv, 0.9 + 0.1 * xf.var(0, unbiased=True),
                          rtol=1e-3)


def test_bn_stats_only_matches_batchnorm():
    from moco_amd import _C
    torch.manual_seed(27)
    R, C = 2048, 64
    x = torch.randn(R, C).cuda().to(torch.bfloat16)
    rm = torch.zeros(C).cuda()
    rv = torch.ones(C).cuda()
    mean, rstd = _C.bn_stats(x, rm, rv, 1e-5, 0.1)
    xf = x.float()
    assert torch.allclose(mean, xf.mean(0), atol=2e-3)
    assert torch.allclose(rstd, torch.rsqrt(xf.var(0, unbiased=False) + 1e-5),
                          rtol=2e-3)
    assert torch.allclose(rm, 0.1 * xf.mean(0), atol=2e-3)


def test_bn_bwd_dx_only_matches_formula():
    from moco_amd import _C
    torch.manual_seed(28)
    R, C = 1024, 64
    gp = (torch.randn(R, C) * 0.1).cuda().to(torch.bfloat16)
    z = torch.randn(R, C).cuda().to(torch.bfloat16)
    mean = torch.randn(C).cuda() * 0.1
    rstd = (torch.rand(C) + 0.5).cuda()
    gamma = (torch.rand(C) + 0.5).cuda()
    dbeta = torch.randn(C).cuda()
    dgamma = torch.randn(C).cuda()
    dx = _C.bn_bwd_dx_only(gp, z, mean, rstd, gamma, dbeta, dgamma)
    zhat = (z.float() - mean[None, :]) * rstd[None, :]
    ref = gamma[None, :] * rstd[None, :] * (
        gp.float() - dbeta[None, :] / R - zhat * dgamma[None, :] / R)
    assert (dx.float() - ref).abs().max() < 2e-2


# ---------------------------------------------------------------------------
# fp8 (e4m3) GEMM on the MX-scaled MFMA (BASELINE config 4)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("M,N,K", [(256, 64, 128), (1536, 128, 256),
                                   (128, 192, 512)])
def test_gemm_nt_fp8_matches_dequant_oracle(M, N, K):
    """The fp8 GEMM must match the fp32 product of the DEQUANTIZED
    operands (same values, fp32 accumulate) to bf16-output rounding."""
    from moco_amd import _C
    torch.manual_seed(50)
    A8 = (torch.randn(M, K) * 100).clamp(-448, 448).cuda() \
        .to(torch.float8_e4m3fn)
    B8 = (torch.randn(N, K) * 100).clamp(-448, 448).cuda() \
        .to(torch.float8_e4m3fn)
    out_scale = 1.0 / (64.0 * 64.0)
    (C,) = _C.gemm_nt_fp8(A8, B8, out_scale)
    ref = (A8.float() @ B8.float().t()) * out_scale
    d = (C.float() - ref).abs()
    tol = ref.abs().max() * 8e-3 + 1e-2
    assert d.max() < tol, (d.max(), ref.abs().max())


def test_gemm_nt_fp8_identity_layout():
    """One-hot rows against an asymmetric B catch fragment transposes."""
    from moco_amd import _C
    M, N, K = 128, 64, 128
    A = torch.zeros(M, K)
    for m in range(M):
        A[m, (m * 7 + 3) % K] = 2.0
    B = (torch.arange(N * K, dtype=torch.float32).view(N, K) % 13) - 6
    A8 = A.cuda().to(torch.float8_e4m3fn)
    B8 = B.cuda().to(torch.float8_e4m3fn)
    (C,) = _C.gemm_nt_fp8(A8, B8, 0.5)
    ref = (A8.float() @ B8.float().t()) * 0.5
    assert torch.allclose(C.float(), ref, atol=1e-2), \
        (C.float() - ref).abs().max()


def test_conv1x1_fp8_drift_vs_bf16():
    """Config-4 drift check: fp8 conv fwd vs the bf16 path on realistic
    (BN-normalized) activations — relative error must stay in the
    low-percent range expected of per-tensor e4m3."""
    from moco_amd.ops.fp8 import conv1x1_fp8_fwd
    torch.manual_seed(51)
    N, Ci, H, Co = 2, 256, 8, 64
    x = torch.randn(N, Ci, H, H).cuda()
    w = (torch.randn(Co, Ci, 1, 1) * 0.05).cuda()
    ref = torch.nn.functional.conv2d(x, w)
    out = conv1x1_fp8_fwd(
        x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last),
        w)
    rel = (out.float() - ref).norm() / ref.norm()
    assert float(rel) < 0.04, float(rel)
    # and strictly more drift than bf16, but same ballpark (sanity)
    assert float(rel) > 1e-4


def test_gemm_nt_fp8_stats_epilogue():
    from moco_amd import _C
    torch.manual_seed(52)
    M, N, K = 256, 64, 128
    A8 = (torch.randn(M, K) * 50).cuda().to(torch.float8_e4m3fn)
    B8 = (torch.randn(N, K) * 50).cuda().to(torch.float8_e4m3fn)
    s = 1.0 / 2500.0
    C, ssum, ssq = _C.gemm_nt_fp8(A8, B8, s, True)
    ref = (A8.float() @ B8.float().t()) * s
    assert torch.allclose(ssum.sum(0), ref.sum(0), rtol=5e-3, atol=0.5)
    assert torch.allclose(ssq.sum(0), (ref ** 2).sum(0), rtol=8e-3, atol=1.0)


def test_head_avgpool_fwd_bwd():
    from moco_amd.ops.pool import head_avgpool
    torch.manual_seed(60)
    for N, C, H in [(2, 2048, 7), (3, 64, 8)]:
        x = (torch.randn(N, C, H, H) * 2).cuda().to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last) \
            .requires_grad_(True)
        y = head_avgpool(x)
        ref = torch.nn.functional.adaptive_avg_pool2d(x.float(), (1, 1))
        assert y.shape == (N, C, 1, 1)
        assert (y.float() - ref).abs().max() < 2e-2
        g = torch.randn_like(y)
        y.backward(g)
        dref = (g.float() / (H * H)).expand(N, C, H, H)
        assert (x.grad.float() - dref).abs().max() < 2e-2

"""CPU-path numerics for the ops layer (these same implementations are the
oracle the GPU kernel tests compare against)."""

import math

import torch

from moco_amd.ops import (contrastive_logits, l2_normalize, queue_enqueue_,
                          softmax_ce_label0)
from moco_amd.ops.multi_tensor import FusedSGD, MultiTensorEMA


def test_l2_normalize_matches_torch():
    x = torch.randn(32, 128, requires_grad=True)
    y = l2_normalize(x, dim=1)
    ref = torch.nn.functional.normalize(x, dim=1)
    assert torch.allclose(y, ref, atol=1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().clone().requires_grad_(True)
    torch.nn.functional.normalize(x2, dim=1).backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)


def test_contrastive_logits_matches_einsum():
    N, C, K = 8, 128, 64
    T = 0.07
    q = torch.randn(N, C, requires_grad=True)
    k = torch.randn(N, C)
    queue_kc = torch.randn(K, C)
    logits = contrastive_logits(q, k, queue_kc, queue_kc.t().contiguous(), T)
    l_pos = torch.einsum("nc,nc->n", q, k).unsqueeze(-1)
    l_neg = torch.einsum("nc,ck->nk", q, queue_kc.t())
    ref = torch.cat([l_pos, l_neg], dim=1) / T
    assert torch.allclose(logits, ref, atol=1e-5)

    g = torch.randn_like(logits)
    logits.backward(g)
    q2 = q.detach().clone().requires_grad_(True)
    l_pos2 = torch.einsum("nc,nc->n", q2, k).unsqueeze(-1)
    l_neg2 = torch.einsum("nc,ck->nk", q2, queue_kc.t())
    (torch.cat([l_pos2, l_neg2], dim=1) / T).backward(g)
    assert torch.allclose(q.grad, q2.grad, atol=1e-4)


def test_softmax_ce_label0_matches_cross_entropy():
    N, K1 = 16, 257
    logits = torch.randn(N, K1, requires_grad=True)
    loss = softmax_ce_label0(logits)
    logits2 = logits.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(
        logits2, torch.zeros(N, dtype=torch.long))
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad, logits2.grad, atol=1e-6)


def test_queue_enqueue_ring():
    C, K, B = 16, 32, 8
    queue = torch.zeros(C, K)
    ptr = torch.zeros(1, dtype=torch.long)
    batches = [torch.randn(B, C) for _ in range(5)]
    for i, keys in enumerate(batches):
        queue_enqueue_(queue, None, ptr, keys)
        assert ptr.item() == ((i + 1) * B) % K
    # after 5 batches of 8 into K=32: columns hold batches [4,1,2,3] rotated
    assert torch.allclose(queue[:, 0:8], batches[4].t())
    assert torch.allclose(queue[:, 8:16], batches[1].t())
    assert torch.allclose(queue[:, 24:32], batches[3].t())


def test_multi_tensor_ema_matches_loop():
    torch.manual_seed(0)
    dst = [torch.randn(7), torch.randn(3, 5), torch.randn(130000)]
    src = [torch.randn_like(t) for t in dst]
    expect = [0.99 * d + 0.01 * s for d, s in zip(dst, src)]
    MultiTensorEMA(dst, src).step(0.99)
    for d, e in zip(dst, expect):
        assert torch.allclose(d, e, atol=1e-6)


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(0)
    p1 = torch.randn(100, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    opt1 = FusedSGD([p1], lr=0.1, momentum=0.9, weight_decay=1e-4)
    opt2 = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for _ in range(5):
        g = torch.randn(100)
        p1.grad = g.clone()
        p2.grad = g.clone()
        opt1.step()
        opt2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_fused_sgd_state_roundtrip():
    p = torch.randn(10, requires_grad=True)
    opt = FusedSGD([p], lr=0.1, momentum=0.9)
    p.grad = torch.randn(10)
    opt.step()
    state = opt.state_dict()
    opt2 = FusedSGD([p], lr=0.1, momentum=0.9)
    opt2.load_state_dict(state)
    b1 = opt.state[p]["momentum_buffer"]
    b2 = opt2.state[p]["momentum_buffer"]
    assert torch.allclose(b1, b2)


def test_bn_channel_support_predicate():
    from moco_amd.ops.bn import _bn_c_supported
    for c in (64, 128, 256, 512, 1024, 2048):
        assert _bn_c_supported(c), c
    for c in (96, 100, 3, 24, 4096, 160):
        assert not _bn_c_supported(c), c


def test_contrastive_odd_dim_cpu_path():
    # dim not divisible by 32: must still work via the fallback math
    q = torch.randn(4, 100, requires_grad=True)
    k = torch.randn(4, 100)
    queue_kc = torch.randn(16, 100)
    logits = contrastive_logits(q, k, queue_kc, queue_kc.t().contiguous(),
                                0.2)
    assert logits.shape == (4, 17)
    logits.sum().backward()
    assert q.grad is not None


def test_fp8_quantize_roundtrip_cpu():
    """Per-tensor e4m3 quantization: dequantized values track the source
    within e4m3's relative precision, amax maps near the format max."""
    import torch

    from moco_amd.ops.fp8 import E4M3_MAX, quantize_e4m3
    torch.manual_seed(3)
    x = torch.randn(64, 128) * 5
    x8, s = quantize_e4m3(x)
    xd = x8.float() * s
    rel = (xd - x).abs() / x.abs().clamp_min(1e-3)
    assert float(rel.median()) < 0.04
    assert float(x8.float().abs().max()) <= E4M3_MAX
    # amax element lands at the top of the representable range
    assert float(x8.float().abs().max()) > 0.9 * E4M3_MAX


def test_eval_transform_batched_matches_single():
    """cpu_two_crop_eval_transform on a [B,3,H,W] batch must equal the
    per-image application (the lincls validate path batches it)."""
    import torch

    from moco_amd.loader import cpu_two_crop_eval_transform
    torch.manual_seed(5)
    tfm = cpu_two_crop_eval_transform(32)
    batch = torch.rand(5, 3, 48, 40)
    out_b = tfm(batch)
    out_s = torch.stack([tfm(batch[i]) for i in range(5)])
    assert out_b.shape == (5, 3, 32, 32)
    assert torch.allclose(out_b, out_s, atol=1e-6)


def test_mirror_set_table_construction():
    """MirrorSet chunk/tile tables (CPU-constructible): every Bottleneck
    1x1 conv gets a [Co,Ci] bf16 mirror + [Ci,Co] transposed mirror, the
    3x3 a channels_last mirror; cast chunks tile the full numel and the
    transpose meta covers every 64x64 tile exactly once."""
    import torch

    from moco_amd.models.resnet import resnet50
    from moco_amd.ops.mirrors import MirrorSet

    m = resnet50(num_classes=8)
    ms = MirrorSet(m)
    # 16 bottlenecks x 3 convs mirrored
    assert len(ms.weights) == 48
    assert len(ms.transposed) == 32  # conv1 + conv3 per block
    for w, mb in ms.weights:
        assert mb.dtype == torch.bfloat16
        assert mb.numel() == w.numel()
    # cast chunks cover every element exactly once
    total = sum(int(s) for s in ms._cast_sz)
    assert total == sum(w.numel() for w, _ in ms.weights)
    # transpose tiles: sum of 64x64 tiles == transposed elements
    assert ms._t_meta.numel() == 4 * ms._t_sp.numel()
    tiles = ms._t_sp.numel()
    assert tiles * 64 * 64 == sum(co * ci for _, _, co, ci in ms.transposed)
    for mb, wt, co, ci in ms.transposed:
        assert wt.shape == (ci, co) and co % 64 == 0 and ci % 64 == 0
    assert ms.valid()


def test_kernel_buckets_tool():
    """The profiling summarizer parses a rocprofv3-style stats CSV."""
    import io
    import subprocess
    import sys

    csv = ('"Name","Calls","TotalDurationNs","AverageNs","Percentage",'
           '"MinNs","MaxNs","StdDev"\n'
           '"gemm_nt_lds_kernel<128, true, true, true>",10,5000000,'
           '500000,50.0,1,2,0\n'
           '"naive_conv_x",2,5000000,2500000,50.0,1,2,0\n')
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".csv",
                                     delete=False) as f:
        f.write(csv)
        path = f.name
    r = subprocess.run([sys.executable, "tools/kernel_buckets.py", path,
                        "2"], capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    assert "our gemm_nt fwd" in r.stdout
    assert "miopen naive_conv" in r.stdout

import math

import pytest
import torch

import moco_amd
from moco_amd.builder import MoCo, moco_loss
from moco_amd.models.resnet import resnet18
from moco_amd.ops import FusedSGD


def make_moco(K=256, mlp=False, T=0.07, m=0.999):
    torch.manual_seed(0)
    return MoCo(lambda num_classes: resnet18(num_classes=num_classes),
                dim=128, K=K, m=m, T=T, mlp=mlp)


def test_logits_shape_and_labels():
    model = make_moco(K=256)
    logits, labels = model(torch.randn(8, 3, 32, 32),
                           torch.randn(8, 3, 32, 32))
    assert logits.shape == (8, 257)
    assert labels.dtype == torch.long and (labels == 0).all()


def test_encoders_start_identical_and_key_frozen():
    model = make_moco()
    for pq, pk in zip(model.encoder_q.parameters(),
                      model.encoder_k.parameters()):
        assert torch.equal(pq, pk)
        assert not pk.requires_grad


def test_queue_init_normalized():
    model = make_moco(K=128)
    norms = model.queue.norm(dim=0)
    assert torch.allclose(norms, torch.ones(128), atol=1e-5)


def test_queue_rotation_and_content():
    model = make_moco(K=64)
    model.eval()  # freeze BN stats; queue mechanics unaffected
    B = 16
    before = model.queue.clone()
    seen_keys = []
    for step in range(5):  # 5*16 = 80 > 64: wraps around
        with torch.no_grad():
            logits, _ = model(torch.randn(B, 3, 32, 32),
                              torch.randn(B, 3, 32, 32))
        assert model.queue_ptr.item() == ((step + 1) * B) % 64
    # after 4 steps the whole queue has been replaced
    assert not torch.allclose(model.queue, before)
    # all columns stay unit-norm (keys are L2-normalized)
    assert torch.allclose(model.queue.norm(dim=0), torch.ones(64), atol=1e-4)


def test_queue_holds_past_keys_not_current():
    """The positives of step t must NOT be in the queue used at step t
    (enqueue happens after the logits: reference builder.py:161)."""
    model = make_moco(K=32)
    model.eval()
    B = 8
    with torch.no_grad():
        im_q = torch.randn(B, 3, 32, 32)
        im_k = torch.randn(B, 3, 32, 32)
        q_before = model.queue.clone()
        logits, _ = model(im_q, im_k, step=0)
        # the logits' negative block must match the *pre-enqueue* queue
        k = model.queue[:, :B].t()  # the keys just enqueued at ptr 0
        # recompute l_neg against the pre-enqueue queue
        # (we can't recover q directly; instead check the queue changed
        #  exactly in the first B columns)
        assert torch.allclose(model.queue[:, B:], q_before[:, B:])
        assert not torch.allclose(model.queue[:, :B], q_before[:, :B])


def test_ema_math():
    model = make_moco(m=0.9)
    pq = list(model.encoder_q.parameters())[0]
    pk = list(model.encoder_k.parameters())[0]
    with torch.no_grad():
        pq.add_(1.0)  # make q differ from k
    k_old = pk.clone()
    q_val = pq.clone()
    model._momentum_update_key_encoder()
    expected = 0.9 * k_old + 0.1 * q_val
    assert torch.allclose(pk, expected, atol=1e-6)


def test_grad_flows_only_through_q():
    model = make_moco(K=64)
    logits, _ = model(torch.randn(4, 3, 32, 32), torch.randn(4, 3, 32, 32))
    moco_loss(logits).backward()
    assert all(p.grad is not None for p in model.encoder_q.parameters())
    assert all(p.grad is None for p in model.encoder_k.parameters())


def test_assert_on_bad_batch_divisibility():
    model = make_moco(K=30)  # 30 % 8 != 0
    with pytest.raises(AssertionError):
        model(torch.randn(8, 3, 32, 32), torch.randn(8, 3, 32, 32))


def test_mlp_head_layout():
    model = make_moco(mlp=True)
    sd = model.state_dict()
    assert "encoder_q.fc.0.weight" in sd and "encoder_q.fc.2.weight" in sd
    assert sd["encoder_q.fc.0.weight"].shape == (512, 512)  # resnet18 width
    assert sd["encoder_q.fc.2.weight"].shape == (128, 512)


def test_gradient_step_decreases_loss():
    """Trainability: one SGD step on encoder_q (key encoder, queue and EMA
    state held fixed) must reduce the InfoNCE on the same inputs."""
    import copy

    torch.manual_seed(0)
    model = make_moco(K=128, T=0.2, mlp=True)
    im_q = torch.rand(8, 3, 32, 32)
    im_k = torch.rand(8, 3, 32, 32)
    snap = copy.deepcopy(model.state_dict())

    logits, _ = model(im_q, im_k, step=0)
    loss1 = moco_loss(logits)
    loss1.backward()
    with torch.no_grad():
        for p in model.encoder_q.parameters():
            p.add_(p.grad, alpha=-1e-3)
    # restore everything except encoder_q
    restore = {k: v for k, v in snap.items()
               if not k.startswith("encoder_q")}
    model.load_state_dict(restore, strict=False)

    logits2, _ = model(im_q, im_k, step=0)
    loss2 = moco_loss(logits2)
    assert float(loss2) < float(loss1), (float(loss1), float(loss2))


def test_shuffle_unshuffle_roundtrip_world1():
    model = make_moco()
    x = torch.randn(16, 3, 8, 8)
    shuffled, idx_unshuffle = model._batch_shuffle(x, step=3)
    restored = model._batch_unshuffle(shuffled, idx_unshuffle)
    assert torch.equal(restored, x)
    assert not torch.equal(shuffled, x)  # permutation actually shuffles


def test_graph_safe_deferred_enqueue_cpu():
    """graph_safe mode: forward defers the enqueue past backward (no queue
    snapshot); flush_queue() applies it.  Semantics on CPU match the eager
    path exactly."""
    m1 = make_moco()
    m2 = make_moco()
    m2.load_state_dict(m1.state_dict())
    m2.graph_safe = True
    x_q = torch.randn(8, 3, 32, 32)
    x_k = torch.randn(8, 3, 32, 32)

    q_before = m2.queue.clone()
    logits1, _ = m1(x_q.clone(), x_k.clone(), step=0)
    logits2, _ = m2(x_q.clone(), x_k.clone(), step=0)
    # deferred: queue untouched until flush
    assert torch.equal(m2.queue, q_before)
    assert m2._pending_keys is not None
    m2.flush_queue()
    assert m2._pending_keys is None
    assert torch.equal(m1.queue, m2.queue)
    assert m1.queue_ptr.item() == m2.queue_ptr.item()
    # world-1 logits identical (shuffle skip is exactly inverted anyway)
    assert torch.allclose(logits1, logits2, atol=1e-5)

"""Property-based tests (hypothesis) for the op layer's CPU oracles —
the same code paths the GPU kernels are validated against."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from moco_amd.ops import contrastive_logits, l2_normalize, softmax_ce_label0
from moco_amd.parallel import step_permutation

small = st.integers(min_value=1, max_value=17)


@settings(max_examples=25, deadline=None)
@given(n=small, c=st.integers(min_value=2, max_value=64),
       seed=st.integers(min_value=0, max_value=10_000))
def test_l2_normalize_unit_rows(n, c, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, c, generator=g) * 3 + 0.1
    y = l2_normalize(x, dim=1)
    norms = y.norm(dim=1)
    assert torch.allclose(norms, torch.ones(n), atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(n=small, k=st.integers(min_value=1, max_value=64),
       seed=st.integers(min_value=0, max_value=10_000))
def test_softmax_ce_bounds_and_grad_sum(n, k, seed):
    g = torch.Generator().manual_seed(seed)
    logits = (torch.randn(n, 1 + k, generator=g) * 5).requires_grad_(True)
    loss = softmax_ce_label0(logits)
    # CE >= 0 always; <= max-logit-gap + log(K+1)
    assert float(loss) >= -1e-5
    loss.backward()
    # softmax-CE gradient rows each sum to zero
    row_sums = logits.grad.sum(dim=1)
    assert torch.allclose(row_sums, torch.zeros(n), atol=1e-6)
    # positive-class grad is non-positive, negatives non-negative
    assert (logits.grad[:, 0] <= 1e-9).all()
    assert (logits.grad[:, 1:] >= -1e-9).all()


@settings(max_examples=20, deadline=None)
@given(n=small, c=st.sampled_from([8, 16, 32]),
       k=st.sampled_from([4, 16, 64]),
       seed=st.integers(min_value=0, max_value=10_000))
def test_contrastive_scale_invariance(n, c, k, seed):
    """Dividing T by 2 must exactly double every logit."""
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(n, c, generator=g)
    kk = torch.randn(n, c, generator=g)
    queue_kc = torch.randn(k, c, generator=g)
    queue_ck = queue_kc.t().contiguous()
    l1 = contrastive_logits(q, kk, queue_kc, queue_ck, 0.2)
    l2 = contrastive_logits(q, kk, queue_kc, queue_ck, 0.1)
    assert torch.allclose(l2, 2 * l1, atol=1e-4)


@settings(max_examples=30, deadline=None)
@given(step=st.integers(min_value=0, max_value=1 << 20),
       batch=st.sampled_from([4, 16, 64, 256]),
       seed=st.integers(min_value=0, max_value=1 << 30))
def test_step_permutation_properties(step, batch, seed):
    idx, inv = step_permutation(step, batch, seed, torch.device("cpu"))
    # a permutation, and inv really inverts it
    assert sorted(idx.tolist()) == list(range(batch))
    assert torch.equal(idx[inv], torch.arange(batch))
    # deterministic
    idx2, _ = step_permutation(step, batch, seed, torch.device("cpu"))
    assert torch.equal(idx, idx2)
    # different steps differ (overwhelmingly likely for batch >= 4)
    idx3, _ = step_permutation(step + 1, batch, seed, torch.device("cpu"))
    if batch >= 16:
        assert not torch.equal(idx, idx3)

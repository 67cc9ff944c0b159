"""Collectives for the MI355X-native MoCo engine (RCCL over xGMI).

Reference collective inventory (upstream builder.py):
- C3/C5: shuffle-BN image/key exchange was an all-gather of the FULL global
  batch on every rank (builder.py:82,112; ~154 MB rebroadcast per step at the
  reference config).  xGMI is a point-to-point mesh (7 links/GPU), so here
  each rank sends only the rows each peer actually needs: an all-to-all-v —
  ~8x less traffic at world size 8.  (gloo has no all_to_all; the CPU
  plumbing path falls back to all-gather + local slice, same semantics.)
- C4: the shuffle permutation was generated on rank 0 and broadcast
  (builder.py:85-88).  Here every rank derives the identical permutation from
  a shared per-step seed (torch.Generator) — zero communication.
- C6: key enqueue all-gather (builder.py:61) stays an all-gather: every rank
  needs every key so the queue stays replicated and deterministic.
"""

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


@torch.no_grad()
def concat_all_gather(tensor: torch.Tensor) -> torch.Tensor:
    """Gather a tensor from all ranks and concat along dim 0 (no autograd),
    parity with reference builder.py:166-178."""
    if not is_distributed():
        return tensor
    world = get_world_size()
    out = [torch.empty_like(tensor) for _ in range(world)]
    dist.all_gather(out, tensor.contiguous())
    return torch.cat(out, dim=0)


def step_permutation(step: int, global_batch: int, seed: int,
                     device: torch.device) -> Tuple[torch.Tensor, torch.Tensor]:
    """The shared per-step shuffle permutation, identical on every rank.

    Returns (idx_shuffle, idx_unshuffle) as int64 CPU tensors (index math is
    host-side; the data exchange itself is the only device traffic).
    """
    g = torch.Generator()
    g.manual_seed((seed * 1_000_003 + step) & 0x7FFFFFFF)
    idx_shuffle = torch.randperm(global_batch, generator=g)
    idx_unshuffle = torch.argsort(idx_shuffle)
    return idx_shuffle, idx_unshuffle


def _backend_supports_a2a() -> bool:
    if not is_distributed():
        return False
    return dist.get_backend() in ("nccl",)  # nccl == RCCL on ROCm


class A2APlan:
    """Host-side plan for one permutation exchange, computed once per step.

    For the shuffle/unshuffle pair the permutation and its inverse are both
    known up front (step_permutation), so neither direction needs the
    O(total log total) argsorts of a from-scratch plan: `dest_perm` (global
    src row -> global dest slot) and `src_of_slot` (its inverse) are passed
    in, and only two length-n argsort/bincounts remain per direction.
    """

    def __init__(self, dest_perm: torch.Tensor, src_of_slot: torch.Tensor,
                 world: int, n: int, rank: int):
        total = world * n
        assert dest_perm.numel() == total
        dest_rank = torch.div(dest_perm, n, rounding_mode="floor")
        my_dest_rank = dest_rank[rank * n:(rank + 1) * n]
        self.send_order = torch.argsort(my_dest_rank, stable=True)
        self.send_splits = torch.bincount(my_dest_rank,
                                          minlength=world).tolist()
        mine_src = src_of_slot[rank * n:(rank + 1) * n]
        src_rank_of_mine = torch.div(mine_src, n, rounding_mode="floor")
        self.recv_splits = torch.bincount(src_rank_of_mine,
                                          minlength=world).tolist()
        # recv rows arrive grouped by source rank; within each group in the
        # sender's send order = ascending global SOURCE id (the stable sort
        # by dest rank keeps rows in local = global source order). So
        # arrival order = my slots sorted by (src_rank, src_global_id):
        order_within = torch.argsort(src_rank_of_mine * total + mine_src)
        pos_of_slot = torch.empty(n, dtype=torch.long)
        pos_of_slot[order_within] = torch.arange(n)
        self.pos_of_slot = pos_of_slot
        self.n = n


@torch.no_grad()
def all_to_all_rows(x: torch.Tensor, dest_perm: torch.Tensor,
                    plan: Optional[A2APlan] = None) -> torch.Tensor:
    """Exchange rows so that output row i on the destination layout equals
    global row dest_perm maps — generic permutation exchange.

    `x`: [N, ...] local rows (global row id = rank*N + i).
    `dest_perm`: [W*N] int64 — dest_perm[g] is the global slot that global
    source row g must land in (slot s lives on rank s // N, position s % N).

    Implemented as all_to_all_single with ragged splits on RCCL; all-gather
    fallback elsewhere.  Every rank computes the same plan from dest_perm.
    """
    world = get_world_size()
    n = x.shape[0]
    if not is_distributed():
        return x[torch.argsort(dest_perm)] if dest_perm is not None else x

    rank = get_rank()

    if not _backend_supports_a2a():
        gathered = concat_all_gather(x)
        # output slot s on this rank holds the source row g with
        # dest_perm[g] == rank*n + s
        src_of_slot = torch.argsort(dest_perm)
        mine = src_of_slot[rank * n:(rank + 1) * n]
        return gathered[mine].contiguous()

    if plan is None:
        plan = A2APlan(dest_perm, torch.argsort(dest_perm), world, n, rank)

    send_buf = x[plan.send_order.to(x.device)].contiguous()
    recv_buf = torch.empty_like(send_buf)
    dist.all_to_all_single(recv_buf, send_buf,
                           output_split_sizes=plan.recv_splits,
                           input_split_sizes=plan.send_splits)
    return recv_buf[plan.pos_of_slot.to(x.device)].contiguous()


class UnshuffleHandle:
    """Opaque handle returned by shuffle_exchange: the unshuffle index plus
    the precomputed reverse A2A plan (so the unshuffle pays no plan cost).
    Index-compatible with the raw idx_unshuffle tensor for tests/tools."""

    def __init__(self, idx_unshuffle: torch.Tensor,
                 plan: Optional[A2APlan] = None):
        self.idx_unshuffle = idx_unshuffle
        self.plan = plan

    def numel(self):
        return self.idx_unshuffle.numel()


@torch.no_grad()
def shuffle_exchange(x: torch.Tensor, step: int, seed: int):
    """Batch-shuffle for shuffling-BN (reference _batch_shuffle_ddp,
    builder.py:76-103).  Returns (shuffled local rows, unshuffle handle)."""
    world = get_world_size()
    n = x.shape[0]
    idx_shuffle, idx_unshuffle = step_permutation(step, world * n, seed,
                                                  x.device)
    if not is_distributed():
        return x[idx_shuffle.to(x.device)], UnshuffleHandle(idx_unshuffle)
    # idx_shuffle maps slot->src, so dest slot of src g is argsort(idx_shuffle)
    # = idx_unshuffle; the inverse map (src_of_slot) is idx_shuffle itself.
    if _backend_supports_a2a():
        rank = get_rank()
        fwd = A2APlan(idx_unshuffle, idx_shuffle, world, n, rank)
        rev = A2APlan(idx_shuffle, idx_unshuffle, world, n, rank)
        out = all_to_all_rows(x, idx_unshuffle, plan=fwd)
        return out, UnshuffleHandle(idx_unshuffle, plan=rev)
    return (all_to_all_rows(x, idx_unshuffle),
            UnshuffleHandle(idx_unshuffle))


@torch.no_grad()
def unshuffle_exchange(x: torch.Tensor, idx_unshuffle) -> torch.Tensor:
    """Inverse of shuffle_exchange (reference _batch_unshuffle_ddp,
    builder.py:106-123).  Accepts the UnshuffleHandle from shuffle_exchange
    (reuses its precomputed plan) or a raw idx_unshuffle tensor."""
    plan = None
    if isinstance(idx_unshuffle, UnshuffleHandle):
        plan = idx_unshuffle.plan
        idx_unshuffle = idx_unshuffle.idx_unshuffle
    if not is_distributed():
        return x[idx_unshuffle.to(x.device)]
    dest_perm = torch.argsort(idx_unshuffle) if plan is None else None
    return all_to_all_rows(x, dest_perm, plan=plan)


@torch.no_grad()
def sync_bn_running_stats(module: torch.nn.Module) -> None:
    """Average BatchNorm running_mean/running_var across ranks.

    DDP is created with broadcast_buffers=False (the queue must stay
    rank-deterministic, and a per-step buffer broadcast is redundant), so
    per-rank BN running stats drift apart over an epoch; call this before
    checkpointing/validation so the saved stats reflect the whole job.
    One flat all-reduce for all BN buffers.
    """
    if not is_distributed():
        return
    bufs = []
    for m in module.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            if m.running_mean is not None:
                bufs.append(m.running_mean)
            if m.running_var is not None:
                bufs.append(m.running_var)
    if not bufs:
        return
    flat = torch.cat([b.detach().reshape(-1) for b in bufs])
    dist.all_reduce(flat, op=dist.ReduceOp.SUM)
    flat.div_(get_world_size())
    off = 0
    for b in bufs:
        n = b.numel()
        b.copy_(flat[off:off + n].view_as(b))
        off += n


@torch.no_grad()
def assert_step_sync(step: int, device: torch.device) -> None:
    """All-gather the step counter and assert every rank agrees.

    Guards the shuffling-BN contract: the shuffle/unshuffle permutation is
    derived from a shared (seed, step) pair, so a single rank skipping a
    step would silently corrupt the key/positive pairing.  Called every
    STEP_SYNC_EVERY steps from MoCo.forward (an 8-byte latency-bound
    collective, amortized to nothing).
    """
    if not is_distributed():
        return
    t = torch.tensor([step], dtype=torch.long, device=device)
    out = [torch.empty_like(t) for _ in range(get_world_size())]
    dist.all_gather(out, t)
    steps = [int(o.item()) for o in out]
    if any(s != step for s in steps):
        raise RuntimeError(
            f"shuffling-BN step counters desynced across ranks: {steps} "
            "(rank-local step drift corrupts the shuffle/unshuffle pairing)")


# ---------------------------------------------------------------------------
# Comm/compute overlap: run the shuffle-BN exchange (C3) on a dedicated HIP
# stream, launched BEFORE the query-encoder forward so the xGMI transfer
# hides under compute (SURVEY §5.8 overlap requirement).  The join point is
# the key-encoder forward.
# ---------------------------------------------------------------------------

_COMM_STREAM = None


def _comm_stream():
    global _COMM_STREAM
    if _COMM_STREAM is None:
        _COMM_STREAM = torch.cuda.Stream()
    return _COMM_STREAM


class _ReadyJob:
    def __init__(self, result):
        self._result = result

    def wait(self):
        return self._result


class _AsyncShuffleJob:
    def __init__(self, out, idx_unshuffle, done_event):
        self._out = out
        self._idx = idx_unshuffle
        self._done = done_event

    def wait(self):
        cur = torch.cuda.current_stream()
        cur.wait_event(self._done)
        # out was allocated on the comm stream; register its use here so the
        # caching allocator doesn't recycle it under the main stream
        self._out.record_stream(cur)
        return self._out, self._idx


def overlap_enabled() -> bool:
    import os
    return os.environ.get("MOCO_AMD_NO_OVERLAP", "0") != "1"


@torch.no_grad()
def begin_shuffle(x: torch.Tensor, step: int, seed: int):
    """Start the shuffle-BN exchange; returns a job whose .wait() yields
    (shuffled rows, idx_unshuffle)."""
    can_overlap = (x.is_cuda and is_distributed()
                   and _backend_supports_a2a() and overlap_enabled())
    if not can_overlap:
        return _ReadyJob(shuffle_exchange(x, step, seed))
    stream = _comm_stream()
    ready = torch.cuda.Event()
    ready.record()  # x is valid once current-stream work is done
    done = torch.cuda.Event()
    with torch.cuda.stream(stream):
        stream.wait_event(ready)
        x.record_stream(stream)  # keep x alive for the comm stream
        out, idx_unshuffle = shuffle_exchange(x, step, seed)
        done.record()
    return _AsyncShuffleJob(out, idx_unshuffle, done)

"""Multi-process distributed tests on gloo/localhost (no GPU) — exercising
the shuffle-BN exchange, gather determinism and queue replication exactly as
they run under RCCL, minus the backend."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from moco_amd import parallel

WORLD = 2


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _run_spawn(fn, port, world=WORLD):
    mp.spawn(fn, args=(world, port), nprocs=world, join=True)


# --- workers (module-level for pickling) -----------------------------------

def _worker_concat_all_gather(rank, world, port):
    _init(rank, world, port)
    x = torch.arange(4, dtype=torch.float32) + 10 * rank
    out = parallel.concat_all_gather(x.unsqueeze(1))
    expect = torch.cat([torch.arange(4, dtype=torch.float32) + 10 * r
                        for r in range(world)]).unsqueeze(1)
    assert torch.equal(out, expect), (rank, out)
    dist.destroy_process_group()


def _worker_shuffle_roundtrip(rank, world, port):
    _init(rank, world, port)
    n = 8
    x = (torch.arange(n, dtype=torch.float32) + 100 * rank).unsqueeze(1) \
        .repeat(1, 3)
    shuffled, idx_unshuffle = parallel.shuffle_exchange(x, step=7, seed=42)
    assert shuffled.shape == x.shape
    restored = parallel.unshuffle_exchange(shuffled, idx_unshuffle)
    assert torch.equal(restored, x), (rank, restored, x)

    # the shuffle is the REFERENCE semantics: global slot s holds global
    # source row idx_shuffle[s]
    idx_shuffle, _ = parallel.step_permutation(7, world * n, 42,
                                               torch.device("cpu"))
    global_rows = torch.cat([
        (torch.arange(n, dtype=torch.float32) + 100 * r).unsqueeze(1)
        .repeat(1, 3) for r in range(world)])
    expect = global_rows[idx_shuffle[rank * n:(rank + 1) * n]]
    assert torch.equal(shuffled, expect), (rank, shuffled, expect)
    dist.destroy_process_group()


def _worker_permutation_identical(rank, world, port):
    _init(rank, world, port)
    idx, inv = parallel.step_permutation(5, 32, 123, torch.device("cpu"))
    gathered = parallel.concat_all_gather(idx.unsqueeze(0).float())
    for r in range(world):
        assert torch.equal(gathered[r].long(), idx), "permutation diverged"
    dist.destroy_process_group()


def _worker_queue_replicated(rank, world, port):
    _init(rank, world, port)
    torch.manual_seed(1234)  # same init on every rank (DDP broadcast analog)
    from moco_amd.builder import MoCo
    from moco_amd.models.resnet import resnet18

    model = MoCo(lambda num_classes: resnet18(num_classes=num_classes),
                 dim=128, K=64, m=0.99, T=0.2)
    model.eval()
    torch.manual_seed(100 + rank)  # different data per rank
    with torch.no_grad():
        for step in range(3):
            model(torch.randn(4, 3, 32, 32), torch.randn(4, 3, 32, 32),
                  step=step)
    # every rank must hold the identical queue (enqueue gathers all keys)
    gathered = parallel.concat_all_gather(model.queue.unsqueeze(0))
    for r in range(world):
        assert torch.allclose(gathered[r], model.queue, atol=1e-6), \
            f"queue diverged on rank {rank} vs {r}"
    assert model.queue_ptr.item() == (3 * 4 * world) % 64
    dist.destroy_process_group()


def _worker_sampler_shards(rank, world, port):
    _init(rank, world, port)
    from torch.utils.data.distributed import DistributedSampler

    from moco_amd.loader import SyntheticImageDataset
    ds = SyntheticImageDataset(length=64, size=8)
    sampler = DistributedSampler(ds, shuffle=True, seed=0)
    sampler.set_epoch(0)
    idxs = torch.tensor(list(iter(sampler)))
    all_idxs = parallel.concat_all_gather(idxs.float().unsqueeze(1))
    uniq = set(all_idxs.long().flatten().tolist())
    assert len(uniq) == 64, "sampler shards must partition the dataset"
    dist.destroy_process_group()


# --- tests ------------------------------------------------------------------

def test_concat_all_gather():
    _run_spawn(_worker_concat_all_gather, 29311)


def test_shuffle_exchange_roundtrip_and_semantics():
    _run_spawn(_worker_shuffle_roundtrip, 29312)


def test_step_permutation_identical_across_ranks():
    _run_spawn(_worker_permutation_identical, 29313)


def test_queue_replicated_across_ranks():
    _run_spawn(_worker_queue_replicated, 29314)


def test_distributed_sampler_shards():
    _run_spawn(_worker_sampler_shards, 29315)


def test_all_to_all_rows_single_process_identity():
    x = torch.randn(6, 3)
    perm = torch.randperm(6)
    # world=1: dest_perm semantics -> output slot s = x[argsort(perm)[s]]
    out = parallel.all_to_all_rows(x, perm)
    assert torch.equal(out, x[torch.argsort(perm)])


def test_shuffle_exchange_world4():
    _run_spawn(_worker_shuffle_roundtrip, 29316, world=4)


def test_queue_replicated_world4():
    _run_spawn(_worker_queue_replicated, 29317, world=4)


def _worker_bench_codepath(rank, world, port):
    """bench.py's exact training-step code path (MoCoStepper + DDP +
    FusedSGD) at gloo world 2: after K steps, params and the queue must be
    bit-identical across ranks (DDP grad sync + deterministic enqueue)."""
    _init(rank, world, port)
    import moco_amd
    from moco_amd.engine import MoCoStepper
    from moco_amd.models.resnet import build_model
    from moco_amd.ops import FusedSGD

    torch.manual_seed(1234 + rank)  # bench.py seeds per rank
    model = moco_amd.MoCo(
        lambda num_classes: build_model("resnet18", num_classes=num_classes),
        dim=128, K=64, m=0.999, T=0.2, mlp=True, shuffle_seed=1234)
    model = torch.nn.parallel.DistributedDataParallel(
        model, broadcast_buffers=False)
    optimizer = FusedSGD(model.parameters(), lr=0.03, momentum=0.9,
                         weight_decay=1e-4)
    stepper = MoCoStepper(model, optimizer, torch.device("cpu"),
                          aug_plus=True, out_size=32,
                          dtype=torch.float32)
    g = torch.Generator().manual_seed(4321 + rank)
    pool = [torch.rand(4, 3, 40, 40, generator=g) for _ in range(2)]
    losses = []
    for i in range(4):
        loss, _, _ = stepper.step_from_raw(pool[i % 2])
        losses.append(float(loss))
    assert all(l == l and l < 100 for l in losses), losses

    raw = model.module
    # queue replicated bit-exactly across ranks
    qs = [torch.empty_like(raw.queue) for _ in range(world)]
    dist.all_gather(qs, raw.queue)
    for r in range(1, world):
        assert torch.equal(qs[0], qs[r]), "queue diverged across ranks"
    assert raw.queue_ptr.item() == (4 * 4 * world) % 64
    # params identical across ranks after DDP-synced updates
    flat = torch.cat([p.detach().reshape(-1)
                      for p in raw.parameters()])[:100000]
    ps = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(ps, flat)
    for r in range(1, world):
        assert torch.equal(ps[0], ps[r]), "params diverged across ranks"
    dist.destroy_process_group()


def test_bench_codepath_world2_numerics():
    _run_spawn(_worker_bench_codepath, 29318)


def _worker_a2a_plan(rank, world, port):
    """A2APlan against the brute-force all-gather+slice fallback for many
    random permutations, and UnshuffleHandle plan reuse."""
    _init(rank, world, port)
    n = 6
    for trial in range(4):
        g = torch.Generator().manual_seed(100 + trial)
        perm = torch.randperm(world * n, generator=g)
        src_of_slot = torch.argsort(perm)
        x = torch.randn(n, 5, generator=torch.Generator().manual_seed(rank))
        plan = parallel.A2APlan(perm, src_of_slot, world, n, rank)
        # brute-force expectation via gather
        allx = parallel.concat_all_gather(x)
        expect = allx[src_of_slot[rank * n:(rank + 1) * n]]
        # gloo has no all_to_all: exercise the plan via its components
        send = x[plan.send_order]
        recvs = parallel.concat_all_gather(send)
        # reconstruct what all_to_all_single would deliver to this rank
        chunks = []
        off = torch.zeros(world, dtype=torch.long)
        sizes = torch.tensor(
            [parallel.A2APlan(perm, src_of_slot, world, n, r).send_splits
             for r in range(world)])
        for src in range(world):
            start = src * n + int(sizes[src][:rank].sum())
            chunks.append(recvs[start:start + int(sizes[src][rank])])
        recv_buf = torch.cat(chunks)
        got = recv_buf[plan.pos_of_slot]
        assert torch.equal(got, expect), (rank, trial)
    dist.destroy_process_group()


def test_a2a_plan_matches_bruteforce():
    _run_spawn(_worker_a2a_plan, 29319)


def test_bench_codepath_world4_numerics():
    _run_spawn(_worker_bench_codepath, 29320, world=4)

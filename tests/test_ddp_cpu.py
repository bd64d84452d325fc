"""Multi-process data-parallel correctness on CPU (gloo, world_size=2),
mirroring the reference's no-GPU distributed mode (SURVEY.md §4.2:
Scaling_model_training.ipynb runs 2-worker DDP with use_gpu=False)."""
import multiprocessing as mp
import os

import pytest
import torch


def _worker(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist

    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.parallel import FlatAdamW, FlatReplica, init_distributed

    init_distributed(backend="gloo")
    torch.manual_seed(42)  # same init on both ranks before broadcast
    cfg = t5_config("flan-t5-small", num_layers=2, num_decoder_layers=2, d_model=64,
                    d_ff=128, num_heads=4, d_kv=16, vocab_size=256, dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    replica = FlatReplica(model, bucket_mb=0.05)  # force several buckets
    opt = FlatAdamW(replica, lr=1e-3, weight_decay=0.01)

    torch.manual_seed(100 + rank)  # different data per rank
    ids = torch.randint(0, 256, (2, 16))
    for _ in range(3):
        opt.zero_grad()
        out = replica(input_ids=ids, labels=ids.clone())
        out.loss.backward()
        replica.finish_grad_sync()
        opt.step()
    # after sync steps the flat params must be identical across ranks
    flat = torch.cat([g.flat_param for _, g in replica.groups]).double()
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    same = torch.allclose(gathered[0], gathered[1], atol=1e-6)
    # single-process equivalent: average both ranks' grads explicitly
    q.put((rank, float(out.loss.item()), bool(same),
           float(flat.sum().item())))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_flat_replica_two_ranks_stay_in_sync():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29531
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    assert all(r[2] for r in results), f"ranks diverged: {results}"
    # both ranks ended with the same params
    assert results[0][3] == pytest.approx(results[1][3], rel=1e-9)


def _worker_equiv(rank, world, port, q):
    """2-rank run must equal a 1-process run on the concatenated batch."""
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.parallel import FlatAdamW, FlatReplica, init_distributed

    init_distributed(backend="gloo")
    torch.manual_seed(7)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1, d_model=32,
                    d_ff=64, num_heads=2, d_kv=16, vocab_size=128, dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    replica = FlatReplica(model, bucket_mb=0.01)
    opt = FlatAdamW(replica, lr=1e-3, weight_decay=0.0, max_grad_norm=None)
    torch.manual_seed(55)
    ids = torch.randint(0, 128, (4, 8))  # same on both ranks
    my = ids[rank * 2:(rank + 1) * 2]
    opt.zero_grad()
    out = replica(input_ids=my, labels=my.clone())
    out.loss.backward()
    replica.finish_grad_sync()
    grad = torch.cat([g.flat_grad for _, g in replica.groups]).clone()
    # plain numpy copy: torch tensors over mp.Queue use fd-based shm reduction
    # that races with child exit
    q.put((rank, grad.numpy().copy()))
    import torch.distributed as dist
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_grad_equals_full_batch_grad():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_equiv, args=(r, 2, 29532, q)) for r in range(2)]
    for p in procs:
        p.start()
    res = {r: torch.from_numpy(g) for r, g in [q.get(timeout=240) for _ in range(2)]}
    for p in procs:
        p.join(timeout=60)

    # single-process full-batch reference
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.parallel import FlatReplica

    torch.manual_seed(7)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1, d_model=32,
                    d_ff=64, num_heads=2, d_kv=16, vocab_size=128, dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    replica = FlatReplica(model, bucket_mb=0.01)
    torch.manual_seed(55)
    ids = torch.randint(0, 128, (4, 8))
    # loss is a mean over tokens; each rank's half has equal token count, so
    # full-batch loss = mean of per-rank losses -> grads average the same way
    l0 = replica(input_ids=ids[:2], labels=ids[:2].clone()).loss
    l1 = replica(input_ids=ids[2:], labels=ids[2:].clone()).loss
    ((l0 + l1) / 2).backward()
    ref = torch.cat([g.flat_grad for _, g in replica.groups])

    assert torch.allclose(res[0], res[1], atol=1e-6)
    assert torch.allclose(res[0].float(), ref.float(), atol=1e-4), \
        (res[0] - ref).abs().max()


@pytest.mark.timeout(300)
def test_flat_replica_four_ranks_stay_in_sync():
    """collective correctness at world sizes beyond 2 —
    4 gloo ranks, bucketed all-reduce, identical params after a step."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29537
    procs = [ctx.Process(target=_worker, args=(r, 4, port, q)) for r in range(4)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(4)]
    for p in procs:
        p.join(timeout=60)
    sums = [r[3] for r in results]
    assert all(s == pytest.approx(sums[0], rel=1e-9) for s in sums), sums


def _worker_explicit(rank, world, port, q):
    """Hook-less backward + explicit all_reduce_grads(): the path the
    two-graph hybrid step uses after a hipGraph backward replay (hooks never
    fire) — must produce the same averaged grads as the hook path."""
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.parallel import FlatReplica, init_distributed

    init_distributed(backend="gloo")
    torch.manual_seed(7)
    cfg = t5_config("flan-t5-small", num_layers=1, num_decoder_layers=1, d_model=32,
                    d_ff=64, num_heads=2, d_kv=16, vocab_size=128, dropout_rate=0.0)
    model = T5ForConditionalGeneration(cfg)
    replica = FlatReplica(model, bucket_mb=0.01)
    for h in replica._hooks:  # emulate graph replay: hooks don't run
        h.remove()
    torch.manual_seed(55)
    ids = torch.randint(0, 128, (4, 8))
    my = ids[rank * 2:(rank + 1) * 2]
    replica.zero_grads()
    out = replica(input_ids=my, labels=my.clone())
    out.loss.backward()
    replica.all_reduce_grads()
    replica.finish_grad_sync()
    grad = torch.cat([g.flat_grad for _, g in replica.groups]).clone()
    q.put((rank, grad.numpy().copy()))
    import torch.distributed as dist
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_explicit_allreduce_matches_hook_path():
    ctx = mp.get_context("spawn")
    q1, q2 = ctx.Queue(), ctx.Queue()
    procs = [ctx.Process(target=_worker_equiv, args=(r, 2, 29541, q1)) for r in range(2)]
    procs += [ctx.Process(target=_worker_explicit, args=(r, 2, 29543, q2)) for r in range(2)]
    for p in procs[:2]:
        p.start()
    hook_grads = {r: g for r, g in (q1.get(timeout=240), q1.get(timeout=240))}
    for p in procs[:2]:
        p.join(timeout=60)
    for p in procs[2:]:
        p.start()
    expl_grads = {r: g for r, g in (q2.get(timeout=240), q2.get(timeout=240))}
    for p in procs[2:]:
        p.join(timeout=60)
    import numpy as np

    for r in (0, 1):
        np.testing.assert_allclose(hook_grads[r], expl_grads[r], rtol=1e-6, atol=1e-7)


def test_deferred_grads_match_inplace():
    """deferred_grads (engine stores + foreach collect) must produce the same
    flat gradients as the pre-assigned in-place accumulation mode."""
    from mi355.models.t5.configuration import t5_config
    from mi355.models.t5.modeling import T5ForConditionalGeneration
    from mi355.parallel import FlatReplica

    def run(deferred):
        torch.manual_seed(7)
        cfg = t5_config("flan-t5-small", num_layers=2, num_decoder_layers=2,
                        d_model=32, d_ff=64, num_heads=2, d_kv=16,
                        vocab_size=128, dropout_rate=0.0)
        model = T5ForConditionalGeneration(cfg)
        replica = FlatReplica(model, deferred_grads=deferred)
        torch.manual_seed(55)
        ids = torch.randint(0, 128, (2, 8))
        for _ in range(2):  # two steps: catches stale-grad accumulation
            replica.zero_grads()
            out = replica(input_ids=ids, labels=ids.clone())
            out.loss.backward()
            replica.collect_grads()
        return torch.cat([g.flat_grad for _, g in replica.groups]).clone()

    a, b = run(False), run(True)
    assert torch.allclose(a, b, atol=1e-7), (a - b).abs().max()


def _worker_mixed_dtype(rank, world, port, q):
    """Mixed-dtype model (bf16 body + fp32 BatchNorm-style params): the
    per-dtype flat groups must broadcast and all-reduce independently."""
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    from mi355.parallel import FlatReplica, init_distributed

    init_distributed(backend="gloo")
    torch.manual_seed(100 + rank)  # DIFFERENT init per rank: broadcast must fix
    model = torch.nn.Sequential(
        torch.nn.Linear(8, 8, bias=False).to(torch.bfloat16),
        torch.nn.Linear(8, 4, bias=True),  # fp32
    )
    replica = FlatReplica(model, bucket_mb=0.001)
    dtypes = sorted(str(g.flat_param.dtype) for _, g in replica.groups)
    x = torch.randn(4, 8, generator=torch.Generator().manual_seed(9))
    replica.zero_grads()
    out = model[1](model[0](x.to(torch.bfloat16)).float())
    out.square().mean().backward()
    replica.finish_grad_sync()
    flat = torch.cat([g.flat_grad.float() for _, g in replica.groups])
    par = torch.cat([g.flat_param.float() for _, g in replica.groups])
    q.put((rank, dtypes, flat.numpy().copy(), par.numpy().copy()))
    import torch.distributed as dist
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_mixed_dtype_groups_sync():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_mixed_dtype, args=(r, 2, 29547, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {r: (d, g, pa) for r, d, g, pa in (q.get(timeout=240), q.get(timeout=240))}
    for p in procs:
        p.join(timeout=60)
    import numpy as np

    # decay-bf16, decay-fp32 (weight), no_decay-fp32 (bias)
    assert res[0][0] == ["torch.bfloat16", "torch.float32", "torch.float32"]
    np.testing.assert_allclose(res[0][2], res[1][2])  # params broadcast from rank0
    np.testing.assert_allclose(res[0][1], res[1][1], rtol=1e-5)  # grads reduced


def test_collect_grads_zeroes_unused_param_slice():
    """Deferred mode: a param whose grad is None this step must have its
    flat-buffer slice ZEROED, not keep last step's gradient (ADVICE r1
    medium)."""
    import torch
    from mi355.parallel.ddp import FlatReplica

    m = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 4))
    rep = FlatReplica(m, deferred_grads=True)
    x = torch.randn(2, 4)
    # step 1: both layers used
    m(x).sum().backward()
    rep.collect_grads()
    rep.finish_grad_sync()
    g0 = None
    for _, g in rep.groups:
        g0 = g.flat_grad.clone()
    rep.zero_grads()
    # step 2: only the first layer used -> layer-2 slices must be zero
    m[0](x).sum().backward()
    rep.collect_grads()
    rep.finish_grad_sync()
    for (_, g), prev in zip(rep.groups, [g0]):
        for p, off in zip(g.params, g.offsets):
            sl = g.flat_grad[off:off + p.numel()]
            if p.grad is None:
                assert torch.all(sl == 0), "stale gradient survived in flat buffer"


def _order_check_worker(rank, world, port, diverge):
    import os

    import torch
    import torch.distributed as dist

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from mi355.parallel.ddp import FlatReplica

    m = torch.nn.Linear(8, 8)
    rep = FlatReplica(m)
    m(torch.randn(2, 8)).sum().backward()
    if diverge and rank == 1:
        # simulate a rank issuing an extra collective
        rep._order_log.append(("bucket", 9, 9, 1))
    try:
        rep.finish_grad_sync()
        ok = True
    except RuntimeError as e:
        ok = "divergence" not in str(e)
    dist.destroy_process_group()
    if diverge and ok:
        raise SystemExit(3)  # should have raised
    if not diverge and not ok:
        raise SystemExit(4)


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.parametrize("diverge", [False, True])
def test_collective_order_divergence_assertion(diverge):
    """every rank must issue the same collectives in the
    same order; a divergence must raise instead of hanging RCCL."""
    import multiprocessing as mp

    port = _free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_order_check_worker, args=(r, 2, port, diverge))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]

"""Multi-process (gloo, world_size=2) tests of the collective layer:
sharded embedding all-to-all lookup/grad, dense allreduce, DLRM step.
Models the reference's in-process multi-task server tests.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from deeprec_amd.embedding.options import InitializerOption


def _run_dist(fn, world_size=2, port=29511):
    ctx = mp.get_context("spawn")
    procs = []
    for r in range(world_size):
        p = ctx.Process(target=_dist_entry,
                        args=(fn.__name__, r, world_size, port))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=120)
    for r, p in enumerate(procs):
        assert p.exitcode == 0, f"rank {r} exited with {p.exitcode}"


def _dist_entry(fn_name, rank, world_size, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        globals()[fn_name](rank, world_size)
    finally:
        if dist.is_initialized():  # live-resize bodies may tear down
            dist.destroy_process_group()


# ---------------- worker bodies ----------------

def _body_sharded_lookup(rank, world):
    from deeprec_amd import EmbeddingVariableOption, RaggedIds
    from deeprec_amd.parallel import (
        ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)

    opt = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=1.0))
    sev = ShardedEmbeddingVariable("sev", 4, ev_option=opt)
    # rank r looks up keys [r+1, 2]; key 2 shared by both ranks
    sp = RaggedIds.from_lists([[rank + 1, 2]])
    out = sharded_embedding_lookup_sparse(sev, sp, combiner="sum")
    assert out.shape == (1, 4)
    torch.testing.assert_close(out, torch.full((1, 4), 2.0))
    # global keys {1, 2}: owner(1)=rank1, owner(2)=rank0 -> one key per shard
    assert sev.size() == 1


def _body_sharded_train(rank, world):
    from deeprec_amd import EmbeddingVariableOption, RaggedIds
    from deeprec_amd.optimizers import GradientDescentOptimizer
    from deeprec_amd.parallel import (
        ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=1.0))
    sev = ShardedEmbeddingVariable("sev_t", 4, ev_option=opt_ev)
    opt = GradientDescentOptimizer(embedding_variables=[sev],
                                   learning_rate=0.1)
    my_key = rank * 10 + 1  # distinct per rank; key 2 shared
    sp = RaggedIds.from_lists([[my_key, 2]])
    out = sharded_embedding_lookup_sparse(sev, sp, combiner="sum")
    out.sum().backward()
    opt.step()
    # key 2 got grad 1 from each of the 2 ranks -> w = 1 - 0.1*2 = 0.8
    # per-rank keys got grad 1 -> w = 0.9
    out2 = sharded_embedding_lookup_sparse(
        sev, RaggedIds.from_lists([[2], [my_key]]), combiner="sum")
    torch.testing.assert_close(out2[0], torch.full((4,), 0.8))
    torch.testing.assert_close(out2[1], torch.full((4,), 0.9))


def _body_dense_allreduce(rank, world):
    from deeprec_amd.parallel import DenseGradAllreducer
    p = torch.nn.Parameter(torch.zeros(10))
    p.grad = torch.full((10,), float(rank + 1))
    red = DenseGradAllreducer([p])
    red.allreduce()
    torch.testing.assert_close(p.grad, torch.full((10,), 1.5))


def _body_dlrm_sharded_step(rank, world):
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdamAsyncOptimizer
    from deeprec_amd.parallel import DenseGradAllreducer, broadcast_parameters

    torch.manual_seed(100 + rank)
    m = DLRM(device="cpu", bf16=False, sharded=True, num_sparse=4)
    broadcast_parameters(m.parameters())
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables(),
                             learning_rate=0.001)
    red = DenseGradAllreducer(m.parameters())
    ds = CriteoSyntheticDataset(batch_size=32, seed=5, rank=rank)
    for step in range(2):
        dense, sparse, labels = ds.next_batch()
        logits = m(dense, sparse[:4])
        loss = m.loss_fn(logits, labels)
        opt.zero_grad()
        loss.backward()
        red.allreduce()
        opt.step()
        assert torch.isfinite(loss)
    # dense params must stay identical across ranks
    flat = torch.cat([p.detach().reshape(-1) for p in m.parameters()])
    flats = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(flats, flat)
    torch.testing.assert_close(flats[0], flats[1])


# ---------------- test entries ----------------

@pytest.mark.parametrize("body,port", [
    (_body_sharded_lookup, 29521),
    (_body_sharded_train, 29522),
    (_body_dense_allreduce, 29523),
    (_body_dlrm_sharded_step, 29524),
])
def test_distributed(body, port):
    _run_dist(body, world_size=2, port=port)


def _body_sharded_collection(rank, world):
    from deeprec_amd import EmbeddingVariableOption
    from deeprec_amd.embedding.collection import EmbeddingCollection
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    def init(t):
        gen = torch.Generator().manual_seed(77)
        t.normal_(0, 1, generator=gen)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4))
    sc = ShardedEmbeddingCollection("sc", ["a", "b"], 8, ev_option=opt_ev)
    # single-process reference collection with identical init
    ref = EmbeddingCollection("ref", ["a", "b"], 8, ev_option=opt_ev)
    torch.testing.assert_close(sc.local.storage.default_values,
                               ref.storage.default_values)
    o_s = AdagradOptimizer(embedding_variables=[sc], learning_rate=0.1)
    o_r = AdagradOptimizer(embedding_variables=[ref], learning_rate=0.1)
    for step in range(3):
        g = torch.Generator().manual_seed(500 + step)
        ids_all = torch.randint(0, 40, (8 * world, 2), generator=g)
        ids_mine = ids_all[rank * 8:(rank + 1) * 8]
        out = sc.lookup_matrix(ids_mine)
        (out ** 2).sum().backward()
        o_s.step()
        # reference trains on the full global batch in one process
        out_r = ref.lookup_matrix(ids_all)
        (out_r ** 2).sum().backward()
        o_r.step()
    # shards together must equal the reference table
    tabs_s = sc.export_tables()
    tabs_r = ref.export_tables()
    for name in ("a", "b"):
        ks, vs, _, _ = tabs_s[name]
        kr, vr, _, _ = tabs_r[name]
        gathered_k = [None] * world
        gathered_v = [None] * world
        dist.all_gather_object(gathered_k, ks)
        dist.all_gather_object(gathered_v, vs)
        ks_all = torch.cat(gathered_k)
        vs_all = torch.cat(gathered_v)
        oi, ri = torch.argsort(ks_all), torch.argsort(kr)
        torch.testing.assert_close(ks_all[oi], kr[ri])
        torch.testing.assert_close(vs_all[oi], vr[ri], rtol=1e-4, atol=1e-5)


def test_sharded_collection():
    _run_dist(_body_sharded_collection, world_size=2, port=29531)


def _body_dlrm_sharded_collection_step(rank, world):
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdamAsyncOptimizer
    from deeprec_amd.parallel import DenseGradAllreducer, broadcast_parameters

    torch.manual_seed(100 + rank)
    m = DLRM(device="cpu", bf16=False, sharded=True, num_sparse=4)
    broadcast_parameters(m.parameters())
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables())
    red = DenseGradAllreducer(m.parameters())
    ds = CriteoSyntheticDataset(batch_size=32, seed=5, rank=rank,
                                matrix_format=True)
    for step in range(2):
        dense, ids, labels = ds.next_batch()
        loss = m.loss_fn(m(dense, ids[:, :4]), labels)
        opt.zero_grad()
        loss.backward()
        red.allreduce()
        opt.step()
        assert torch.isfinite(loss)


def test_dlrm_sharded_collection():
    _run_dist(_body_dlrm_sharded_collection_step, world_size=2, port=29532)


def _body_sharded_collection_w4(rank, world):
    """world_size=4 shard-merge equivalence (closer to the 8-GPU shape)."""
    from deeprec_amd import EmbeddingVariableOption
    from deeprec_amd.embedding.collection import EmbeddingCollection
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    def init(t):
        gen = torch.Generator().manual_seed(9)
        t.normal_(0, 1, generator=gen)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4))
    sc = ShardedEmbeddingCollection("sc4", ["a", "b", "c"], 4,
                                    ev_option=opt_ev)
    ref = EmbeddingCollection("ref4", ["a", "b", "c"], 4, ev_option=opt_ev)
    o_s = AdagradOptimizer(embedding_variables=[sc], learning_rate=0.1)
    o_r = AdagradOptimizer(embedding_variables=[ref], learning_rate=0.1)
    for step in range(2):
        g = torch.Generator().manual_seed(800 + step)
        ids_all = torch.randint(0, 60, (4 * world, 3), generator=g)
        out = sc.lookup_matrix(ids_all[rank * 4:(rank + 1) * 4])
        (out ** 2).sum().backward()
        o_s.step()
        (ref.lookup_matrix(ids_all) ** 2).sum().backward()
        o_r.step()
    tabs_s, tabs_r = sc.export_tables(), ref.export_tables()
    for name in tabs_s:
        ks, vs, _, _ = tabs_s[name]
        kr, vr, _, _ = tabs_r[name]
        gk = [None] * world
        gv = [None] * world
        dist.all_gather_object(gk, ks)
        dist.all_gather_object(gv, vs)
        ka, va = torch.cat(gk), torch.cat(gv)
        oi, ri = torch.argsort(ka), torch.argsort(kr)
        torch.testing.assert_close(ka[oi], kr[ri])
        torch.testing.assert_close(va[oi], vr[ri], rtol=1e-4, atol=1e-5)


def test_sharded_collection_world4():
    _run_dist(_body_sharded_collection_w4, world_size=4, port=29541)


def _body_work_queue_dist(rank, world):
    from deeprec_amd.data.parquet import WorkQueue
    wq = WorkQueue([f"f{i}" for i in range(5)])
    first = wq.take()
    assert first == f"f{rank}"
    second = wq.take()
    expected = rank + world
    assert second == (f"f{expected}" if expected < 5 else None)


def test_work_queue_distributed():
    _run_dist(_body_work_queue_dist, world_size=2, port=29542)


def _body_work_queue_tail_drain(rank, world):
    """Odd item count: one rank draws None in the TAIL round while the
    peer still drew an item. The drained rank must keep participating
    in assignment rounds until ALL ranks see the end — the naive
    return-None-and-stop protocol deadlocked the peer's next take()."""
    from deeprec_amd.data.parquet import WorkQueue
    wq = WorkQueue([f"f{i}" for i in range(3)])
    got = []
    while True:
        item = wq.take()
        if item is None:
            break
        got.append(item)
    gathered = [None, None]
    dist.all_gather_object(gathered, got)
    assert sorted(gathered[0] + gathered[1]) == ["f0", "f1", "f2"]
    # both ranks reached None (this line executing on both IS the
    # no-deadlock proof); a further collective round still agrees
    assert wq.take() is None


def test_work_queue_tail_drain():
    _run_dist(_body_work_queue_tail_drain, world_size=2, port=29547)


def _body_bf16_exchange(rank, world):
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    torch.manual_seed(7)
    names = [f"t{i}" for i in range(4)]
    ids = torch.randint(0, 50, (16, 4))
    g1 = torch.Generator().manual_seed(5)
    g2 = torch.Generator().manual_seed(5)
    sev32 = ShardedEmbeddingCollection("x32", names, 8, generator=g1)
    sev16 = ShardedEmbeddingCollection("x16", names, 8, generator=g2,
                                       comm_dtype=torch.bfloat16)
    out32 = sev32.lookup_matrix(ids, train=True)
    out16 = sev16.lookup_matrix(ids, train=True)
    # identical math up to the bf16 transport quantization of the rows
    torch.testing.assert_close(out16, out32, rtol=1e-2, atol=1e-2)
    out16.sum().backward()
    dist.barrier()


def test_bf16_row_exchange_world2():
    _run_dist(_body_bf16_exchange, world_size=2, port=29543)


def _body_async_allreduce(rank, world):
    from deeprec_amd.parallel import DenseGradAllreducer

    torch.manual_seed(rank)
    p = torch.nn.Parameter(torch.randn(1000))
    p.grad = torch.full((1000,), float(rank + 1))
    red = DenseGradAllreducer([p])
    red.allreduce(async_op=True)
    red.wait()
    # mean of (1, 2) = 1.5 for world=2
    expect = sum(range(1, world + 1)) / world
    assert torch.allclose(p.grad, torch.full((1000,), expect))
    dist.barrier()


def test_async_allreduce_world2():
    _run_dist(_body_async_allreduce, world_size=2, port=29544)


def _body_hierarchical_a2a(rank, world):
    from deeprec_amd.parallel import comm
    from deeprec_amd.parallel.hierarchical import hierarchical_all_to_all

    torch.manual_seed(100 + rank)
    for trial in range(3):
        splits = torch.randint(0, 7, (world,)).tolist()
        if trial == 2:
            splits[rank] = 0  # empty self-block edge case
        rows = torch.randn(sum(splits), 5) + rank * 100
        flat_counts = comm.exchange_counts(
            torch.tensor(splits, dtype=torch.int64))
        flat = comm.all_to_all_single(rows, splits, flat_counts.tolist())
        hier, hc = hierarchical_all_to_all(rows.clone(), splits,
                                           node_size=2)
        assert hc == flat_counts.tolist()
        torch.testing.assert_close(hier, flat)
    dist.barrier()


def test_hierarchical_a2a_world4():
    _run_dist(_body_hierarchical_a2a, world_size=4, port=29545)


def _body_sharded_hierarchical(rank, world):
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    torch.manual_seed(3)
    names = [f"t{i}" for i in range(4)]
    ids = torch.randint(0, 60, (16, 4))
    g1 = torch.Generator().manual_seed(9)
    g2 = torch.Generator().manual_seed(9)
    flat = ShardedEmbeddingCollection("hflat", names, 8, generator=g1)
    hier = ShardedEmbeddingCollection("hhier", names, 8, generator=g2,
                                      node_size=2)
    o1 = flat.lookup_matrix(ids, train=True)
    o2 = hier.lookup_matrix(ids, train=True)
    torch.testing.assert_close(o1, o2)
    (o1.sum() + o2.sum()).backward()
    dist.barrier()


def test_sharded_hierarchical_world4():
    _run_dist(_body_sharded_hierarchical, world_size=4, port=29546)


def _body_elastic_resize(rank, world):
    """world=2 trains, coordinator proposes shrink to 1, both checkpoint;
    the restore-under-new-world path is covered by the repartition tests."""
    import tempfile

    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.embedding import EmbeddingVariable, embedding_lookup
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel.elastic import (ElasticAgent,
                                              ElasticController,
                                              elastic_step_hook)

    port = 29650
    ctrl = ElasticController(port=port) if rank == 0 else None
    dist.barrier()  # store server up before agents connect
    agent = ElasticAgent(port=port)

    torch.manual_seed(rank)
    ev = EmbeddingVariable(f"elastic_{rank}", 4, device="cpu")
    opt = AdagradOptimizer(embedding_variables=[ev], learning_rate=0.1)
    saver = Saver(embedding_variables=[ev], rank=rank, world_size=world)
    tmp = tempfile.mkdtemp(prefix=f"elastic{rank}_")

    stopped_at = None
    for step in range(6):
        ids = torch.randint(0, 40, (8,))
        out = embedding_lookup(ev, ids, train=True)
        (out ** 2).sum().backward()
        opt.step()
        if rank == 0 and step == 2:
            ctrl.propose_resize(1)
        dist.barrier()  # all ranks see the event at the same boundary
        if elastic_step_hook(agent, saver, tmp, global_step=step):
            stopped_at = step
            break
    assert stopped_at == 2  # the step the event was proposed
    import glob as g
    assert g.glob(tmp + "/ckpt-*"), "scale event must leave a checkpoint"
    dist.barrier()


def test_elastic_resize_world2():
    _run_dist(_body_elastic_resize, world_size=2, port=29547)


def _body_padded_a2a(rank, world):
    from deeprec_amd.parallel import comm

    torch.manual_seed(20 + rank)
    splits = [rank + 1, 3 - rank][:world]
    rows = torch.randn(sum(splits), 4) + rank * 10
    flat_counts = comm.exchange_counts(
        torch.tensor(splits, dtype=torch.int64))
    flat = comm.all_to_all_single(rows, splits, flat_counts.tolist())
    cap = 8
    padded, counts = comm.padded_all_to_all(rows, splits, cap)
    assert padded.shape[0] == world * cap
    assert counts.tolist() == flat_counts.tolist()
    trimmed = torch.cat([padded[p * cap: p * cap + int(counts[p])]
                         for p in range(world)])
    torch.testing.assert_close(trimmed, flat)
    # pad region is zeros on the wire
    for p in range(world):
        tail = padded[p * cap + int(counts[p]): (p + 1) * cap]
        assert torch.equal(tail, torch.zeros_like(tail))
    # over-cap raises
    try:
        comm.padded_all_to_all(rows, [cap + 1] + [0] * (world - 1), cap)
        raise AssertionError("expected ValueError")
    except ValueError:
        pass
    dist.barrier()


def test_padded_a2a_world2():
    _run_dist(_body_padded_a2a, world_size=2, port=29548)


def test_hierarchical_a2a_world6():
    """3 nodes x 2 GPUs — non-power-of-two node count."""
    _run_dist(_body_hierarchical_a2a, world_size=6, port=29549)


def _body_padded_collection(rank, world):
    """static_mode (padded fixed-shape exchange) must train identically to
    the single-process reference collection — the wire protocol that the
    captured distributed step replays on hardware."""
    from deeprec_amd import EmbeddingVariableOption
    from deeprec_amd.embedding.collection import EmbeddingCollection
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    def init(t):
        gen = torch.Generator().manual_seed(21)
        t.normal_(0, 1, generator=gen)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4))
    sc = ShardedEmbeddingCollection("pc", ["a", "b"], 8, ev_option=opt_ev)
    sc.static_mode(pad_cap=64)
    ref = EmbeddingCollection("pref", ["a", "b"], 8, ev_option=opt_ev)
    o_s = AdagradOptimizer(embedding_variables=[sc], learning_rate=0.1)
    o_r = AdagradOptimizer(embedding_variables=[ref], learning_rate=0.1)
    for step in range(3):
        g = torch.Generator().manual_seed(900 + step)
        ids_all = torch.randint(0, 40, (8 * world, 2), generator=g)
        out = sc.lookup_matrix(ids_all[rank * 8:(rank + 1) * 8])
        (out ** 2).sum().backward()
        o_s.step()
        (ref.lookup_matrix(ids_all) ** 2).sum().backward()
        o_r.step()
    tabs_s, tabs_r = sc.export_tables(), ref.export_tables()
    for name in ("a", "b"):
        ks, vs, fs, _ = tabs_s[name]
        kr, vr, fr, _ = tabs_r[name]
        gk = [None] * world
        gv = [None] * world
        gf = [None] * world
        dist.all_gather_object(gk, ks)
        dist.all_gather_object(gv, vs)
        dist.all_gather_object(gf, fs)
        ka, va, fa = torch.cat(gk), torch.cat(gv), torch.cat(gf)
        oi, ri = torch.argsort(ka), torch.argsort(kr)
        torch.testing.assert_close(ka[oi], kr[ri])
        torch.testing.assert_close(va[oi], vr[ri], rtol=1e-4, atol=1e-5)
        # frequency counters ride the wire (summed true counts)
        torch.testing.assert_close(fa[oi], fr[ri])


def test_padded_collection_world2():
    _run_dist(_body_padded_collection, world_size=2, port=29552)


def test_padded_collection_world4():
    _run_dist(_body_padded_collection, world_size=4, port=29553)


def _body_padded_overflow(rank, world):
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    sc = ShardedEmbeddingCollection("povf", ["a"], 4)
    sc.static_mode(pad_cap=2)  # far below the per-peer unique count
    ids = torch.arange(32).reshape(32, 1)
    try:
        out = sc.lookup_matrix(ids)
        out.sum().backward()
        raise AssertionError("expected pad-cap overflow ValueError")
    except ValueError as e:
        assert "overflow" in str(e)
    dist.barrier()


def test_padded_overflow_world2():
    _run_dist(_body_padded_overflow, world_size=2, port=29554)


def _body_padded_bf16_wire(rank, world):
    """Padded exchange with bf16 row transport matches fp32 within bf16
    quantization (the gloo int16-view wire branch)."""
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    names = [f"t{i}" for i in range(3)]
    ids = torch.randint(0, 50, (16, 3),
                        generator=torch.Generator().manual_seed(4))
    g1 = torch.Generator().manual_seed(6)
    g2 = torch.Generator().manual_seed(6)
    s32 = ShardedEmbeddingCollection("pw32", names, 8, generator=g1)
    s16 = ShardedEmbeddingCollection("pw16", names, 8, generator=g2,
                                     comm_dtype=torch.bfloat16)
    s32.static_mode(pad_cap=128)
    s16.static_mode(pad_cap=128)
    o32 = s32.lookup_matrix(ids, train=True)
    o16 = s16.lookup_matrix(ids, train=True)
    torch.testing.assert_close(o16, o32, rtol=1e-2, atol=1e-2)
    (o16.sum() + o32.sum()).backward()
    dist.barrier()


def test_padded_bf16_wire_world2():
    _run_dist(_body_padded_bf16_wire, world_size=2, port=29555)


def _body_sharded_ev_checkpoint(rank, world):
    """Sharded plain-EV save/restore, including 64-bit-hash ids (>= 2^48
    and negative): restore ownership must match the full-key % world
    routing that lookups use (regression: ADVICE r1 #2/#3)."""
    import tempfile

    from deeprec_amd import EmbeddingVariableOption, RaggedIds
    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.embedding.options import InitializerOption
    from deeprec_amd.optimizers import GradientDescentOptimizer
    from deeprec_amd.parallel import (
        ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=1.0))
    sev = ShardedEmbeddingVariable("sev_ck", 4, ev_option=opt_ev)
    opt = GradientDescentOptimizer(embedding_variables=[sev],
                                   learning_rate=0.1)
    # ids beyond 2^48 and negative exercise full-key routing
    big, neg = (1 << 50) + 3, -7
    sp = RaggedIds.from_lists([[big, neg, 2]])
    out = sharded_embedding_lookup_sparse(sev, sp, combiner="sum")
    out.sum().backward()
    opt.step()
    tmp = [None]
    if rank == 0:
        tmp[0] = tempfile.mkdtemp(prefix="sevck_")
    dist.broadcast_object_list(tmp, src=0)
    saver = Saver(embedding_variables=[sev], rank=rank, world_size=world)
    saver.save(tmp[0], global_step=1)
    dist.barrier()
    # fresh shards (same logical name -> same part files) restore
    sev2 = ShardedEmbeddingVariable("sev_ck", 4, ev_option=opt_ev)
    saver2 = Saver(embedding_variables=[sev2], rank=rank, world_size=world)
    import glob as g
    ckpt = g.glob(tmp[0] + "/ckpt-*")[0]
    saver2.restore(ckpt)
    out2 = sharded_embedding_lookup_sparse(
        sev2, RaggedIds.from_lists([[big], [neg], [2]]), combiner="sum")
    # every key got grad 1 from each of the 2 ranks -> 1 - 0.1*2 = 0.8;
    # a wrong restore ownership would return the 1.0 default instead
    torch.testing.assert_close(out2, torch.full((3, 4), 0.8))
    dist.barrier()


def test_sharded_ev_checkpoint_world2():
    _run_dist(_body_sharded_ev_checkpoint, world_size=2, port=29550)


def _body_sharded_ev_eval_no_insert(rank, world):
    """train=False sharded lookup must not insert keys, bump metadata, or
    build autograd state (regression: ADVICE r1 #4)."""
    from deeprec_amd import EmbeddingVariableOption, RaggedIds
    from deeprec_amd.embedding.options import InitializerOption
    from deeprec_amd.parallel import (
        ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=2.0))
    sev = ShardedEmbeddingVariable("sev_ev", 4, ev_option=opt_ev)
    sp = RaggedIds.from_lists([[5, 6]])
    out = sharded_embedding_lookup_sparse(sev, sp, combiner="sum",
                                          train=False)
    # default rows come back, nothing was inserted anywhere
    torch.testing.assert_close(out, torch.full((1, 4), 4.0))
    assert not out.requires_grad
    assert sev.size() == 0
    sizes = [None] * world
    dist.all_gather_object(sizes, sev.size())
    assert all(s == 0 for s in sizes)
    dist.barrier()


def test_sharded_ev_eval_world2():
    _run_dist(_body_sharded_ev_eval_no_insert, world_size=2, port=29551)


def _body_live_resize(rank, world):
    """IN-PROCESS shrink 3 -> 2: shards re-gather and re-route with
    optimizer state intact; the departed rank's keys live on. Continued
    training matches a single-process reference exactly."""
    import os as _os

    from deeprec_amd import EmbeddingVariableOption
    from deeprec_amd.embedding import EmbeddingVariable, embedding_lookup
    from deeprec_amd.embedding.options import InitializerOption
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel import comm
    from deeprec_amd.parallel.elastic import live_resize
    from deeprec_amd.parallel.sharded_embedding import (
        ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)
    from deeprec_amd import RaggedIds

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=1.0))
    sev = ShardedEmbeddingVariable("lrs", 4, ev_option=opt_ev)
    opt = AdagradOptimizer(embedding_variables=[sev], learning_rate=0.1)
    # single-process oracle replaying the same global batches
    oracle = EmbeddingVariable("lrs_oracle", 4, ev_option=opt_ev)
    opt_o = AdagradOptimizer(embedding_variables=[oracle],
                             learning_rate=0.1)

    def global_step(step, active_world):
        ids = torch.arange(step * 7, step * 7 + 12, dtype=torch.int64)
        out = sharded_embedding_lookup_sparse(
            sev, RaggedIds.from_lists([ids.tolist()]), combiner="sum")
        out.sum().backward()
        opt.step()
        # the owner applies ONE grad summed over the active ranks, so
        # the oracle applies the same total once (optimizers are
        # nonlinear — sequential applies would diverge)
        o = embedding_lookup(oracle, ids, train=True)
        (o * float(active_world)).sum().backward()
        opt_o.step()

    for step in range(2):
        global_step(step, world)

    def reinit(r, w):
        _os.environ["MASTER_PORT"] = "29583"
        dist.init_process_group("gloo", rank=r, world_size=w)

    survived = live_resize(2, [sev], reinit)
    assert survived == (rank < 2)
    if not survived:
        return  # departed rank exits; its shard moved to the survivors
    # shards must now cover ALL keys under the new routing
    for step in range(2, 4):
        global_step(step, 2)
    k, v, f, _ = sev.export()
    ks = [None, None]
    vs = [None, None]
    dist.all_gather_object(ks, k)
    dist.all_gather_object(vs, v)
    ka, va = torch.cat(ks), torch.cat(vs)
    ko, vo, _, _ = oracle.export()
    oi, ri = torch.argsort(ka), torch.argsort(ko)
    torch.testing.assert_close(ka[oi], ko[ri])
    torch.testing.assert_close(va[oi], vo[ri], rtol=1e-5, atol=1e-6)


def test_live_resize_world3_to_2():
    _run_dist(_body_live_resize, world_size=3, port=29582)

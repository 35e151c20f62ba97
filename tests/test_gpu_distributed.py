"""World=1 RCCL tests on a single GPU.

The nccl/RCCL-specific branches (dist.all_to_all_single, bf16 wire dtype,
device-resident split handling, collectives under hipGraph capture) run
here through a world_size=1 process group — the exact code path an 8-GPU
node runs, self-exchange included, so the first multi-GPU run measures
instead of debugging (reference protocol: SOK two-phase exchange,
all2all_input_dispatcher.cu:250-280)."""
import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(scope="module")
def nccl_world1():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29563")
    created = False
    if not dist.is_initialized():
        torch.cuda.set_device(0)
        dist.init_process_group("nccl", rank=0, world_size=1)
        created = True
    yield
    if created:
        # the module captured collectives into hipGraphs — a clean
        # destroy waits forever on their never-completing work records
        from deeprec_amd.parallel import comm
        comm.shutdown(after_capture=True)


def test_rccl_all_to_all_world1(nccl_world1):
    """RCCL self-exchange for every wire dtype the engine ships."""
    from deeprec_amd.parallel import comm

    assert comm.world_size() == 1 and comm.is_initialized()
    for dtype in (torch.float32, torch.bfloat16, torch.int64, torch.int32):
        x = (torch.arange(24, device=DEV).reshape(8, 3) * 7).to(dtype)
        out = comm.all_to_all_single(x, [8], [8])
        assert torch.equal(out, x)
    counts = comm.exchange_counts(
        torch.tensor([5], dtype=torch.int64, device=DEV))
    assert counts.cpu().tolist() == [5]


def test_padded_sharded_world1_matches_local(nccl_world1):
    """Eager padded exchange (HIP dedup/route kernels + RCCL a2a) must
    train identically to the plain local collection."""
    from deeprec_amd.embedding.collection import EmbeddingCollection
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               InitializerOption)
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    def init(t):
        g = torch.Generator().manual_seed(31)
        t.normal_(0, 1, generator=g)

    opt_ev = EmbeddingVariableOption(
        init_option=InitializerOption(initializer=init, default_value_dim=4),
        init_capacity=1 << 14)
    sc = ShardedEmbeddingCollection("w1pad", ["a", "b", "c"], 16,
                                    ev_option=opt_ev, device=DEV)
    ref = EmbeddingCollection("w1ref", ["a", "b", "c"], 16,
                              ev_option=opt_ev, device=DEV)
    sc.enable_graph_mode(expected_entries=1 << 14,
                         expected_slots=1 << 14, pad_cap=2048)
    o_s = AdagradOptimizer(embedding_variables=[sc], learning_rate=0.1)
    o_r = AdagradOptimizer(embedding_variables=[ref], learning_rate=0.1)
    for step in range(4):
        g = torch.Generator().manual_seed(600 + step)
        ids = torch.randint(0, 500, (64, 3), generator=g).to(DEV)
        out_s = sc.lookup_matrix(ids)
        out_r = ref.lookup_matrix(ids)
        torch.testing.assert_close(out_s, out_r, rtol=1e-5, atol=1e-5)
        (out_s ** 2).sum().backward()
        (out_r ** 2).sum().backward()
        o_s.step()
        o_r.step()
    sc.storage._check_error()
    tabs_s, tabs_r = sc.export_tables(), ref.export_tables()
    for name in ("a", "b", "c"):
        ks, vs, fs, _ = tabs_s[name]
        kr, vr, fr, _ = tabs_r[name]
        # PAD_KEY is engine-internal and must NOT leak into exports
        oi, ri = torch.argsort(ks), torch.argsort(kr)
        torch.testing.assert_close(ks[oi], kr[ri])
        torch.testing.assert_close(fs[oi], fr[ri])
        torch.testing.assert_close(vs[oi], vr[ri], rtol=1e-4, atol=1e-5)


def test_padded_bf16_wire_world1(nccl_world1):
    """bf16 row transport over RCCL (ncclBfloat16): quantization-level
    agreement with the fp32 wire."""
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    ids = torch.randint(0, 300, (32, 4),
                        generator=torch.Generator().manual_seed(8)).to(DEV)
    g1 = torch.Generator().manual_seed(14)
    g2 = torch.Generator().manual_seed(14)
    s32 = ShardedEmbeddingCollection("w1w32", [f"t{i}" for i in range(4)],
                                     8, device=DEV, generator=g1)
    s16 = ShardedEmbeddingCollection("w1w16", [f"t{i}" for i in range(4)],
                                     8, device=DEV, generator=g2,
                                     comm_dtype=torch.bfloat16)
    s32.enable_graph_mode(1 << 14, 1 << 14, pad_cap=1024)
    s16.enable_graph_mode(1 << 14, 1 << 14, pad_cap=1024)
    o32 = s32.lookup_matrix(ids)
    o16 = s16.lookup_matrix(ids)
    torch.testing.assert_close(o16.float(), o32.float(),
                               rtol=1e-2, atol=1e-2)
    (o16.sum() + o32.sum()).backward()
    s16.storage._check_error()


def test_captured_padded_step_replay_world1(nccl_world1):
    """The FULL distributed step — requester dedup, padded RCCL
    all-to-alls, owner admission, pooled fwd/bwd, grad exchange, fused
    AdamAsync apply — captured in ONE hipGraph and replayed over fresh
    id batches; trained tables must match the same steps run eagerly."""
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               InitializerOption)
    from deeprec_amd.optimizers import AdamAsyncOptimizer
    from deeprec_amd.parallel.sharded_collection import (
        ShardedEmbeddingCollection)

    def init(t):
        g = torch.Generator().manual_seed(51)
        t.normal_(0, 1, generator=g)

    def make(name):
        opt_ev = EmbeddingVariableOption(
            init_option=InitializerOption(initializer=init,
                                          default_value_dim=4),
            init_capacity=1 << 14)
        sc = ShardedEmbeddingCollection(name, ["a", "b"], 16,
                                        ev_option=opt_ev, device=DEV)
        sc.enable_graph_mode(1 << 14, 1 << 14, pad_cap=2048)
        sc.get_slab("adam_m", 16, 0.0)
        sc.get_slab("adam_v", 16, 0.0)
        opt = AdamAsyncOptimizer(embedding_variables=[sc],
                                 learning_rate=0.01, graph_safe=True)
        return sc, opt

    batches = [torch.randint(0, 400, (64, 2),
                             generator=torch.Generator().manual_seed(i)
                             ).to(DEV) for i in range(7)]

    def eager_step(sc, opt, ids):
        out = sc.lookup_matrix(ids)
        opt.zero_grad()
        (out ** 2).sum().backward()
        opt.step()

    sg, og = make("w1cap")
    # warm 2 steps eagerly (padded path, RCCL comms established)
    eager_step(sg, og, batches[0])
    eager_step(sg, og, batches[1])
    torch.cuda.synchronize()
    sids = batches[2].clone()
    graph = torch.cuda.CUDAGraph()
    og.zero_grad()
    with torch.cuda.graph(graph):
        out = sg.lookup_matrix(sids)
        (out ** 2).sum().backward()
        og.step()
    torch.cuda.synchronize()
    for b in batches[3:7]:
        sids.copy_(b)
        graph.replay()
    torch.cuda.synchronize()
    sg.storage._check_error()

    # eager reference over the batches the graph model actually trained on
    # (capture records batch[2] without executing it)
    se, oe = make("w1eag")
    for b in batches[:2] + batches[3:7]:
        eager_step(se, oe, b)
    se.storage._check_error()

    tg, te = sg.export_tables(), se.export_tables()
    for name in ("a", "b"):
        kg, vg, fg, _ = tg[name]
        ke, ve, fe, _ = te[name]
        gi, ei = torch.argsort(kg), torch.argsort(ke)
        torch.testing.assert_close(kg[gi], ke[ei])
        torch.testing.assert_close(fg[gi], fe[ei])
        torch.testing.assert_close(vg[gi], ve[ei], rtol=1e-3, atol=1e-4)


def test_dense_allreduce_capture_world1(nccl_world1):
    """Bucketed dense all-reduce (async + wait) inside a hipGraph."""
    from deeprec_amd.parallel import DenseGradAllreducer

    p = torch.nn.Parameter(torch.randn(4096, device=DEV))
    p.grad = torch.zeros_like(p)
    red = DenseGradAllreducer([p])
    src = torch.randn(4096, device=DEV)
    # warm: establish the communicator outside capture
    p.grad.copy_(src)
    red.allreduce(async_op=True)
    red.wait()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        p.grad.copy_(src)
        red.allreduce(async_op=True)
        red.wait()
    torch.cuda.synchronize()
    src.copy_(torch.ones(4096, device=DEV) * 3.0)
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(p.grad, torch.full((4096,), 3.0, device=DEV))

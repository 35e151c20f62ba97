"""Property-based fuzzing of pure-CPU subsystems against simple oracles
(hypothesis; the reference validates the same surfaces with hand-picked
cases — random op sequences catch the interleavings hand-written tests
miss)."""
import pytest
import torch

hyp = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings  # noqa: E402
from hypothesis import strategies as st  # noqa: E402


# ---------------- SsdKv vs dict oracle ----------------

_ops = st.lists(
    st.tuples(st.sampled_from(["write", "delete", "compact"]),
              st.lists(st.integers(0, 30), min_size=1, max_size=8)),
    min_size=1, max_size=30)


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(ops=_ops)
def test_ssd_kv_matches_dict_oracle(tmp_path_factory, ops):
    """Random write/delete/compact sequences: SsdKv must always read
    back exactly what a dict would (values keyed by latest write),
    report the same membership, and never resurrect deleted keys."""
    from deeprec_amd.embedding.ssd_kv import SsdKv
    d = tmp_path_factory.mktemp("ssd")
    kv = SsdKv(str(d), dim=4, file_capacity_rows=16)
    oracle = {}
    serial = 0
    for op, keys in ops:
        k = torch.tensor(sorted(set(keys)), dtype=torch.int64)
        if op == "write":
            serial += 1
            vals = torch.full((k.numel(), 4), float(serial))
            vals[:, 0] = k.float()  # per-key distinguishable rows
            kv.write(k, vals)
            for i, key in enumerate(k.tolist()):
                oracle[key] = vals[i]
        elif op == "delete":
            kv.delete(k)
            for key in k.tolist():
                oracle.pop(key, None)
        else:
            kv.compact(sync=True)
        assert kv.size() == len(oracle)
    probe = torch.arange(0, 31, dtype=torch.int64)
    mask = kv.contains(probe)
    assert mask.tolist() == [int(i) in oracle for i in range(31)]
    live = probe[mask]
    if live.numel():
        got = kv.read(live)
        want = torch.stack([oracle[int(i)] for i in live.tolist()])
        torch.testing.assert_close(got, want)


# ---------------- Kafka wire round-trip ----------------

@settings(max_examples=30, deadline=None)
@given(msgs=st.lists(st.binary(min_size=0, max_size=200), min_size=1,
                     max_size=20),
       start=st.integers(0, 5))
def test_kafka_message_set_roundtrip(msgs, start):
    """encode_message_set -> decode_message_set is identity for any
    payload bytes (CRC'd v0 framing, data/kafka_wire.py)."""
    from deeprec_amd.data.kafka_wire import (decode_message_set,
                                             encode_message_set)
    enc = encode_message_set([(start + i, m) for i, m in enumerate(msgs)])
    out = decode_message_set(enc)
    assert [m for _, m in out] == msgs
    assert [o for o, _ in out] == list(range(start, start + len(msgs)))


# ---------------- RESP2 round-trip ----------------

@settings(max_examples=20, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(kvs=st.dictionaries(
    st.text(alphabet=st.characters(min_codepoint=33, max_codepoint=126),
            min_size=1, max_size=12),
    st.binary(min_size=0, max_size=64), min_size=1, max_size=12))
def test_resp2_wire_roundtrip(redis_pair, kvs):
    """Arbitrary binary values survive SET/GET/MGET through the real
    RESP2 wire (serving/redis_store.py client + in-process server)."""
    client = redis_pair
    for k, v in kvs.items():
        assert client.execute("SET", k, v) in (b"OK", "OK")
    for k, v in kvs.items():
        assert client.execute("GET", k) == v
    keys = list(kvs)
    got = client.execute("MGET", *keys, "missing-key-xyzzy")
    assert got[:-1] == [kvs[k] for k in keys]
    assert got[-1] is None


@pytest.fixture(scope="module")
def redis_pair():
    from deeprec_amd.serving.redis_store import MiniRedisServer, RedisClient
    srv = MiniRedisServer()
    client = RedisClient("127.0.0.1", srv.port)
    yield client
    client.close()
    srv.close()


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(n=st.integers(1, 40), dim=st.integers(1, 16),
       seed=st.integers(0, 1 << 20))
def test_redis_feature_store_tensor_roundtrip(redis_store, n, dim, seed):
    """Tensor rows survive the feature-store put/get framing for any
    (rows, dim) shape; missing keys come back as the default fill."""
    store = redis_store
    g = torch.Generator().manual_seed(seed)
    keys = torch.randperm(1 << 16, generator=g)[:n]
    vals = torch.randn(n, dim, generator=g)
    table = f"t{dim}_{seed % 7}"
    store.put(table, keys, vals)
    got = store.get(table, keys, dim)
    torch.testing.assert_close(got, vals)
    miss = store.get(table, torch.tensor([(1 << 20) + 7]), dim,
                     default=3.5)
    assert bool((miss == 3.5).all())


@pytest.fixture(scope="module")
def redis_store():
    from deeprec_amd.serving.redis_store import (MiniRedisServer,
                                                 RedisFeatureStore)
    srv = MiniRedisServer()
    store = RedisFeatureStore("127.0.0.1", srv.port)
    yield store
    srv.close()

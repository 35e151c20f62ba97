"""Append-only SSD KV store: round trips, overwrite semantics,
compaction (sync + async), crash-safe reopen (reference capability:
ssd_hash_kv.h / emb_file.h, tested in embedding_variable_ops_test.cc)."""
import torch

from deeprec_amd.embedding.ssd_kv import SsdKv


def test_roundtrip_and_overwrite(tmp_path):
    kv = SsdKv(str(tmp_path / "s"), dim=8, file_capacity_rows=64)
    keys = torch.arange(100, dtype=torch.int64)
    vals = torch.randn(100, 8)
    kv.write(keys, vals)
    assert kv.size() == 100
    got = kv.read(keys)
    torch.testing.assert_close(got, vals)
    # overwrite half with new values — append + dead-mark, reads follow
    vals2 = torch.randn(50, 8)
    kv.write(keys[:50], vals2)
    torch.testing.assert_close(kv.read(keys[:50]), vals2)
    torch.testing.assert_close(kv.read(keys[50:]), vals[50:])
    # unknown keys read as default
    missing = kv.read(torch.tensor([10_000]), default=0.0)
    assert torch.equal(missing, torch.zeros(1, 8))
    assert kv.contains(torch.tensor([0, 10_000])).tolist() == [True, False]


def test_compaction_sync(tmp_path):
    kv = SsdKv(str(tmp_path / "s"), dim=4, file_capacity_rows=32,
               compact_live_ratio=0.6)
    keys = torch.arange(128, dtype=torch.int64)
    kv.write(keys, torch.ones(128, 4))
    files_before = kv.file_count()
    # overwrite 3/4 of the keys -> old files mostly dead
    kv.write(keys[:96], torch.full((96, 4), 2.0))
    n = kv.compact(sync=True)
    assert n > 0
    assert kv.file_count() < files_before + 4  # dead files reclaimed
    torch.testing.assert_close(kv.read(keys[:96]),
                               torch.full((96, 4), 2.0))
    torch.testing.assert_close(kv.read(keys[96:]), torch.ones(32, 4))


def test_compaction_async(tmp_path):
    kv = SsdKv(str(tmp_path / "s"), dim=4, file_capacity_rows=16,
               compact_live_ratio=0.9)
    keys = torch.arange(64, dtype=torch.int64)
    kv.write(keys, torch.randn(64, 4))
    ref = kv.read(keys)
    kv.write(keys[:48], ref[:48] * 3)
    kv.compact(sync=False)
    kv.wait_compaction()
    torch.testing.assert_close(kv.read(keys[:48]), ref[:48] * 3)
    torch.testing.assert_close(kv.read(keys[48:]), ref[48:])


def test_reopen(tmp_path):
    p = str(tmp_path / "s")
    kv = SsdKv(p, dim=8, file_capacity_rows=32)
    keys = torch.arange(50, dtype=torch.int64) * 7
    vals = torch.randn(50, 8)
    kv.write(keys, vals)
    kv.write(keys[:10], vals[:10] + 1)
    del kv
    kv2 = SsdKv(p, dim=8, file_capacity_rows=32)
    assert kv2.size() == 50
    torch.testing.assert_close(kv2.read(keys[:10]), vals[:10] + 1)
    torch.testing.assert_close(kv2.read(keys[10:]), vals[10:])
    # compaction still works after reopen
    kv2.write(keys, vals - 5)
    kv2.compact(sync=True)
    torch.testing.assert_close(kv2.read(keys), vals - 5)


def test_delete_and_compact(tmp_path):
    kv = SsdKv(str(tmp_path / "s"), dim=4, file_capacity_rows=16,
               compact_live_ratio=0.99)
    keys = torch.arange(48, dtype=torch.int64)
    kv.write(keys, torch.randn(48, 4))
    kv.delete(keys[:40])
    assert kv.size() == 8
    kv.compact(sync=True)
    assert kv.size() == 8
    assert not bool(kv.contains(keys[:1])[0])

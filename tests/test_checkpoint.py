"""Full + incremental checkpoint round-trip tests (modeled on the
reference's incr_ckpt_test.py and EV export/import coverage)."""
import os

import torch

from deeprec_amd import EmbeddingVariable, RaggedIds, embedding_lookup_sparse
from deeprec_amd.checkpoint.saver import Saver, latest_checkpoint
from deeprec_amd.embedding.collection import EmbeddingCollection
from deeprec_amd.embedding.variable import GLOBAL_STEP
from deeprec_amd.optimizers import AdagradOptimizer, AdamAsyncOptimizer


def _train_steps(model_lin, ev, opt, steps, seed=0):
    torch.manual_seed(seed)
    for _ in range(steps):
        sp = RaggedIds.from_lists(
            [torch.randint(0, 30, (2,)).tolist() for _ in range(8)])
        emb = embedding_lookup_sparse(ev, sp, combiner="sum")
        loss = model_lin(emb).pow(2).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()


def test_full_checkpoint_roundtrip(tmp_path):
    lin = torch.nn.Linear(4, 2)
    ev = EmbeddingVariable("ck_ev", 4)
    opt = AdagradOptimizer(params=lin.parameters(),
                           embedding_variables=[ev], learning_rate=0.1)
    saver = Saver(module=lin, embedding_variables=[ev], optimizer=opt)
    _train_steps(lin, ev, opt, 5)
    path = saver.save(str(tmp_path), GLOBAL_STEP.value)
    assert latest_checkpoint(str(tmp_path)) == path

    # continue training to diverge state, then restore and compare
    w_saved = lin.weight.detach().clone()
    keys, vals_saved, freqs_saved, _ = ev.export()
    _train_steps(lin, ev, opt, 3, seed=1)
    assert not torch.allclose(lin.weight.detach(), w_saved)

    lin2 = torch.nn.Linear(4, 2)
    ev2 = EmbeddingVariable("ck_ev2", 4)
    opt2 = AdagradOptimizer(params=lin2.parameters(),
                            embedding_variables=[ev2], learning_rate=0.1)
    saver2 = Saver(module=lin2, embedding_variables=[ev2], optimizer=opt2)
    # restore uses the saved EV file name: rename lookup via evs list order
    saver2.evs = [ev2]
    ev2.name = "ck_ev"
    step = saver2.restore(path)
    assert step == 5
    torch.testing.assert_close(lin2.weight.detach(), w_saved)
    k2, v2, f2, _ = ev2.export()
    o1, o2 = torch.argsort(keys), torch.argsort(k2)
    torch.testing.assert_close(vals_saved[o1], v2[o2])
    torch.testing.assert_close(freqs_saved[o1], f2[o2])
    # optimizer slab restored -> identical continued training
    _train_steps(lin2, ev2, opt2, 3, seed=1)


def test_incremental_checkpoint(tmp_path):
    lin = torch.nn.Linear(4, 1)
    ev = EmbeddingVariable("incr_ev", 4)
    opt = AdamAsyncOptimizer(params=lin.parameters(),
                             embedding_variables=[ev], learning_rate=0.01)
    saver = Saver(module=lin, embedding_variables=[ev], optimizer=opt)
    _train_steps(lin, ev, opt, 4)
    saver.save(str(tmp_path), GLOBAL_STEP.value)
    # more training touches more keys -> incremental save
    _train_steps(lin, ev, opt, 4, seed=2)
    saver.incremental_save(str(tmp_path), GLOBAL_STEP.value)
    keys_final, vals_final, *_ = ev.export()

    lin2 = torch.nn.Linear(4, 1)
    ev2 = EmbeddingVariable("incr_ev_r", 4)
    ev2.name = "incr_ev"
    opt2 = AdamAsyncOptimizer(params=lin2.parameters(),
                              embedding_variables=[ev2], learning_rate=0.01)
    saver2 = Saver(module=lin2, embedding_variables=[ev2], optimizer=opt2)
    step = saver2.restore(latest_checkpoint(str(tmp_path)))
    assert step == 8  # incr checkpoint replayed on top of the full one
    k2, v2, *_ = ev2.export()
    o1, o2 = torch.argsort(keys_final), torch.argsort(k2)
    torch.testing.assert_close(keys_final[o1], k2[o2])
    torch.testing.assert_close(vals_final[o1], v2[o2])


def test_collection_checkpoint_roundtrip(tmp_path):
    coll = EmbeddingCollection("ck_coll", ["a", "b"], 4)
    opt = AdagradOptimizer(embedding_variables=[coll], learning_rate=0.1)
    ids = torch.tensor([[1, 2], [3, 4]])
    out = coll.lookup_matrix(ids)
    out.sum().backward()
    opt.step()
    saver = Saver(embedding_variables=[coll], optimizer=opt)
    path = saver.save(str(tmp_path), 1)

    coll2 = EmbeddingCollection("ck_coll", ["a", "b"], 4)
    saver2 = Saver(embedding_variables=[coll2])
    saver2.restore(path)
    out1 = coll.lookup_matrix(ids, train=False)
    out2 = coll2.lookup_matrix(ids, train=False)
    torch.testing.assert_close(out1, out2)


def test_keep_checkpoint_max(tmp_path):
    ev = EmbeddingVariable("ck_max", 4)
    saver = Saver(embedding_variables=[ev], keep_checkpoint_max=2)
    for s in range(5):
        GLOBAL_STEP.value = s
        saver.save(str(tmp_path), s)
    remaining = sorted(os.listdir(tmp_path))
    assert len([d for d in remaining if not d.endswith(".incr")]) == 2


def test_monitored_session_with_hooks(tmp_path):
    from deeprec_amd.training.session import (
        LoggingTensorHook, MonitoredTrainingSession, StepCounterHook)
    lin = torch.nn.Linear(4, 1)
    ev = EmbeddingVariable("sess_ev", 4)
    opt = AdagradOptimizer(params=lin.parameters(),
                           embedding_variables=[ev], learning_rate=0.1)
    saver = Saver(module=lin, embedding_variables=[ev], optimizer=opt)

    def step_fn():
        sp = RaggedIds.from_lists([[1, 2]])
        loss = lin(embedding_lookup_sparse(ev, sp, combiner="sum")).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()
        return {"loss": loss.detach()}

    with MonitoredTrainingSession(
            hooks=[LoggingTensorHook(2), StepCounterHook(2)],
            checkpoint_dir=str(tmp_path), saver=saver,
            save_checkpoint_steps=3, max_steps=7) as sess:
        while not sess.should_stop():
            sess.run(step_fn)
    assert GLOBAL_STEP.value == 7
    assert latest_checkpoint(str(tmp_path)) is not None

    # resume: a fresh session restores and continues to a later stop step
    ev2 = EmbeddingVariable("sess_ev2", 4)
    ev2.name = "sess_ev"
    lin2 = torch.nn.Linear(4, 1)
    opt2 = AdagradOptimizer(params=lin2.parameters(),
                            embedding_variables=[ev2], learning_rate=0.1)
    saver2 = Saver(module=lin2, embedding_variables=[ev2], optimizer=opt2)

    def step_fn2():
        sp = RaggedIds.from_lists([[1, 2]])
        loss = lin2(embedding_lookup_sparse(ev2, sp, combiner="sum")).sum()
        opt2.zero_grad()
        loss.backward()
        opt2.step()
        return {"loss": loss.detach()}

    with MonitoredTrainingSession(checkpoint_dir=str(tmp_path),
                                  saver=saver2, max_steps=10) as sess:
        start = GLOBAL_STEP.value
        assert start == 7
        while not sess.should_stop():
            sess.run(step_fn2)
    assert GLOBAL_STEP.value == 10


def test_estimator_train_eval(tmp_path):
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdagradOptimizer
    from deeprec_amd.training.estimator import Estimator, RunConfig

    def model_fn(params):
        m = DLRM(device="cpu", bf16=False)
        opt = AdagradOptimizer(params=m.parameters(),
                               embedding_variables=m.embedding_variables(),
                               learning_rate=0.05)
        return m, opt

    def input_fn():
        ds = CriteoSyntheticDataset(batch_size=32, seed=1,
                                    matrix_format=True)
        return iter(ds)

    est = Estimator(model_fn, model_dir=str(tmp_path),
                    config=RunConfig(save_checkpoints_steps=4,
                                     log_step_count_steps=5))
    est.train(input_fn, steps=6, micro_batch=2)
    metrics = est.evaluate(input_fn, steps=2)
    assert "loss" in metrics and metrics["loss"] > 0
    from deeprec_amd.checkpoint.saver import latest_checkpoint
    assert latest_checkpoint(str(tmp_path)) is not None
    preds = next(est.predict(input_fn))
    assert preds.shape == (32,)


def test_repartitioned_restore(tmp_path):
    """Elastic capability: a checkpoint written by 2 shards restores into
    1 (and vice versa) — the reference's 1000-bucket repartition-safe
    scheme (Embedding-Variable-Export-Format.md)."""
    from deeprec_amd.checkpoint.saver import Saver

    # simulate two ranks' shard files: keys split by key % 2
    ev_full = EmbeddingVariable("repart_ev", 4)
    embedding_lookup_sparse(
        ev_full, RaggedIds.from_dense(torch.arange(100).unsqueeze(1)),
        combiner="sum")
    keys, values, freqs, versions = ev_full.export()
    import os

    from safetensors.torch import save_file
    path = os.path.join(str(tmp_path), "ckpt-1")
    os.makedirs(path)
    for r in range(2):
        mask = (keys % 2) == r
        save_file({"keys": keys[mask], "values": values[mask],
                   "freqs": freqs[mask], "versions": versions[mask],
                   "buckets": (keys[mask] % 1000).to(torch.int32)},
                  os.path.join(path, f"ev-repart_ev-part{r}.safetensors"))
    import json
    json.dump({"global_step": 1},
              open(os.path.join(path, "checkpoint.json"), "w"))

    # restore the 2-shard checkpoint into a single EV
    ev1 = EmbeddingVariable("repart_restored", 4)
    ev1.name = "repart_ev"
    saver = Saver(embedding_variables=[ev1])
    saver.restore(path)
    k1, v1, *_ = ev1.export()
    o0, o1 = torch.argsort(keys), torch.argsort(k1)
    torch.testing.assert_close(keys[o0], k1[o1])
    torch.testing.assert_close(values[o0], v1[o1])


def test_cleanup_prunes_fulls_and_stale_incrementals(tmp_path):
    """keep_checkpoint_max prunes old fulls AND incremental deltas older
    than the oldest kept full (they can never be replayed: restore =
    latest full + NEWER deltas only)."""
    import glob
    import os

    import torch
    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.embedding import EmbeddingVariable

    ev = EmbeddingVariable("prune/ev", 4)
    saver = Saver(embedding_variables=[ev], keep_checkpoint_max=2)
    for step in range(1, 6):
        ev.lookup_or_create(torch.tensor([step, step + 100]))
        saver.save(str(tmp_path), step * 10)
        ev.lookup_or_create(torch.tensor([step + 200]))
        saver.incremental_save(str(tmp_path), step * 10 + 5)
    fulls = sorted(os.path.basename(p) for p in
                   glob.glob(str(tmp_path / "ckpt-*"))
                   if not p.endswith(".incr"))
    incrs = sorted(os.path.basename(p) for p in
                   glob.glob(str(tmp_path / "ckpt-*.incr")))
    assert fulls == ["ckpt-40", "ckpt-50"]
    # deltas older than ckpt-40 are gone; newer ones kept
    assert incrs == ["ckpt-45.incr", "ckpt-55.incr"]

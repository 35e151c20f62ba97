"""Estimator facade.

Capability parity with the reference's tf.estimator surface used by the
modelzoo (train/evaluate/predict loops over a model_fn, model_dir
checkpointing, RunConfig). The eager-first contract:

    def model_fn(params) -> (model, optimizer)
    def input_fn() -> iterable of (features..., labels)

Estimator wires MonitoredTrainingSession + Saver + hooks around them.
Micro-batch gradient accumulation (reference: micro_batch_num,
graph_execution_state.cc:635-727 graph-duplication) maps to N sub-batch
backward passes with merged sparse grads before one optimizer step.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Optional

import torch

from deeprec_amd.checkpoint.saver import Saver
from deeprec_amd.training.session import (
    LoggingTensorHook, MonitoredTrainingSession, StepCounterHook)


@dataclass
class RunConfig:
    save_checkpoints_steps: Optional[int] = None
    save_checkpoints_secs: Optional[float] = None
    save_incremental_checkpoint_secs: Optional[float] = None
    keep_checkpoint_max: int = 5
    log_step_count_steps: int = 100


class Estimator:
    def __init__(self, model_fn: Callable, model_dir: str = None,
                 config: Optional[RunConfig] = None, params: dict = None):
        self.model_dir = model_dir
        self.config = config or RunConfig()
        self.params = params or {}
        self.model, self.optimizer = model_fn(self.params)
        self.saver = Saver(
            module=self.model,
            embedding_variables=self.model.embedding_variables(),
            optimizer=self.optimizer,
            keep_checkpoint_max=self.config.keep_checkpoint_max)

    def _step(self, batch, micro_batch: int = 1):
        *features, labels = batch
        self.optimizer.zero_grad()
        if micro_batch <= 1:
            logits = self.model(*features)
            loss = self.model.loss_fn(logits, labels)
            loss.backward()
        else:
            total = 0.0
            bsz = labels.shape[0] // micro_batch
            for i in range(micro_batch):
                sl = slice(i * bsz, (i + 1) * bsz)
                f_i = [x[sl] if torch.is_tensor(x) else x for x in features]
                lg = self.model(*f_i)
                sub = self.model.loss_fn(lg, labels[sl]) / micro_batch
                sub.backward()
                total += float(sub)
            loss = torch.tensor(total)
        self.optimizer.step()
        return {"loss": loss.detach() if torch.is_tensor(loss) else loss}

    def train(self, input_fn: Callable, steps: Optional[int] = None,
              max_steps: Optional[int] = None, hooks=None,
              micro_batch: int = 1):
        it = iter(input_fn())
        from deeprec_amd.embedding.variable import get_global_step
        stop_at = max_steps or (get_global_step() + (steps or 1))
        all_hooks = list(hooks or []) + [
            LoggingTensorHook(self.config.log_step_count_steps),
            StepCounterHook(self.config.log_step_count_steps)]
        with MonitoredTrainingSession(
                hooks=all_hooks, checkpoint_dir=self.model_dir,
                saver=self.saver if self.model_dir else None,
                save_checkpoint_steps=self.config.save_checkpoints_steps,
                save_checkpoint_secs=self.config.save_checkpoints_secs,
                save_incremental_checkpoint_secs=(
                    self.config.save_incremental_checkpoint_secs),
                max_steps=stop_at) as sess:
            while not sess.should_stop():
                sess.run(lambda: self._step(next(it), micro_batch))
        return self

    @torch.no_grad()
    def evaluate(self, input_fn: Callable, steps: int = 10) -> dict:
        """loss / accuracy / streaming AUC over `steps` eval batches
        (reference: modelzoo train.py eval loops report loss+acc+auc)."""
        from deeprec_amd.training.metrics import StreamingAUC
        it = iter(input_fn())
        auc = StreamingAUC()
        losses, n_correct, n_total = [], 0, 0
        for _ in range(steps):
            *features, labels = next(it)
            logits = self.model(*features, train=False)
            if isinstance(logits, (list, tuple)):
                logits = logits[0]
            losses.append(float(self.model.loss_fn(logits, labels)))
            probs = torch.sigmoid(logits)
            auc.update(probs, labels)
            preds = (probs > 0.5).float()
            n_correct += int((preds == labels).sum())
            n_total += labels.numel()
        return {"loss": sum(losses) / len(losses),
                "accuracy": n_correct / max(n_total, 1),
                "auc": auc.result()}

    def train_and_evaluate(self, train_input_fn: Callable,
                           eval_input_fn: Callable, train_steps: int,
                           eval_steps: int = 10,
                           eval_every: Optional[int] = None,
                           hooks=None) -> list:
        """Interleaved train/eval (tf.estimator.train_and_evaluate
        surface): trains `train_steps` total, evaluating every
        `eval_every` steps (default: once at the end). Returns the list
        of eval metric dicts in order."""
        results = []
        every = eval_every or train_steps
        done = 0
        while done < train_steps:
            chunk = min(every, train_steps - done)
            self.train(train_input_fn, steps=chunk, hooks=hooks)
            done += chunk
            results.append(self.evaluate(eval_input_fn, steps=eval_steps))
        return results

    @classmethod
    def run_cluster(cls, model_fn: Callable, input_fn: Callable,
                    tables: dict, tf_config=None, steps: int = 100,
                    model_dir: str = None, config=None, params=None,
                    ps_optimizer: str = "adagrad", ps_lr: float = 0.1):
        """The multi-node Estimator story (reference: TF_CONFIG +
        ClusterSpec + replica_device_setter, modelzoo train.py:858-913):
        the SAME entry runs every task.

        - task.type == "ps": host the EV shards behind the pull/push
          plane (training/cluster.start_ps) and serve until killed;
        - task.type in ("worker", "chief"): build the model via
          model_fn(params) where params["embeddings"] maps table name ->
          PsShardedEmbedding, then run the normal Estimator train loop
          (sparse grads ride the PS pushes inside backward).
        """
        from deeprec_amd.training.cluster import (parse_tf_config,
                                                  start_ps,
                                                  worker_embeddings)
        cfg = parse_tf_config(tf_config)
        if cfg["type"] == "ps":
            server = start_ps(cfg, tables, optimizer=ps_optimizer,
                              lr=ps_lr, checkpoint_dir=model_dir)
            server._thread.join()  # serve until the process is killed
            return None
        p = dict(params or {})
        p["embeddings"] = worker_embeddings(cfg, tables)
        p["cluster"] = cfg
        est = cls(model_fn, model_dir=model_dir, config=config, params=p)
        if cfg["type"] == "evaluator":
            # evaluator task: read-only pulls (no inserts, no pushes),
            # reports loss/accuracy/AUC against the live PS shards
            metrics = est.evaluate(input_fn, steps=steps)
            est.eval_metrics = metrics
            return est
        est.train(input_fn, steps=steps)
        for emb in p["embeddings"].values():
            emb.flush()
        if model_dir and int(cfg.get("index", 0)) == 0:
            # the chief checkpoints every PS shard at the end of its run
            # — the failover artifact a restarted PS restores from
            # (reference: async-PS failover, Incremental-Checkpoint.md)
            client = next(iter(p["embeddings"].values())).client
            for i in range(client.world):
                client.call(i, {"op": "SAVE", "dir": model_dir,
                                "step": steps})
        return est

    @torch.no_grad()
    def predict(self, input_fn: Callable):
        for batch in input_fn():
            features = batch[:-1] if isinstance(batch, tuple) else batch
            logits = self.model(*features, train=False)
            if isinstance(logits, (list, tuple)):
                yield [torch.sigmoid(lg) for lg in logits]
            else:
                yield torch.sigmoid(logits)

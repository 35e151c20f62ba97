"""Model-zoo runner CLI end-to-end on CPU (train.py-analog flag surface)."""
from deeprec_amd.models.runner import main


def test_runner_with_filters_and_ckpt(tmp_path):
    main(["--model", "deepfm", "--steps", "3", "--batch_size", "64",
          "--ev_filter", "counter", "--filter_freq", "1",
          "--ev_elimination", "gstep", "--no_bf16",
          "--checkpoint_dir", str(tmp_path), "--save_steps", "2",
          "--log_steps", "2", "--no_smartstaged"])
    import os
    assert any(d.startswith("ckpt-") for d in os.listdir(tmp_path))


def test_runner_sequence_model():
    main(["--model", "din", "--steps", "2", "--batch_size", "32",
          "--no_bf16", "--optimizer", "adagrad"])


def test_runner_micro_batch_and_parquet(tmp_path):
    import subprocess
    import sys

    import numpy as np
    import pandas as pd

    df = pd.DataFrame(
        {"label": np.random.randint(0, 2, 128).astype("float32")})
    for i in range(13):
        df[f"dense_{i}"] = np.random.randn(128).astype("float32")
    for i in range(26):
        df[f"sparse_{i}"] = np.random.randint(0, 50, 128)
    fn = str(tmp_path / "t.parquet")
    df.to_parquet(fn)
    r = subprocess.run(
        [sys.executable, "-m", "deeprec_amd.models.runner", "--model",
         "dlrm", "--steps", "2", "--batch_size", "32", "--no_bf16",
         "--micro_batch", "2", "--parquet", fn],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0 and "RESULT" in r.stdout, r.stderr[-500:]


def test_runner_workqueue_and_parity_flags(tmp_path):
    """--workqueue shards parquet files through the checkpointable
    WorkQueue and stops cleanly when the queue drains; the reference
    always-on flags (--ev --emb_fusion --op_fusion --group_embedding)
    are accepted (modelzoo train.py CLI parity)."""
    import subprocess
    import sys

    import numpy as np
    import pandas as pd

    for part in range(2):
        df = pd.DataFrame(
            {"label": np.random.randint(0, 2, 64).astype("float32")})
        for i in range(13):
            df[f"dense_{i}"] = np.random.randn(64).astype("float32")
        for i in range(26):
            df[f"sparse_{i}"] = np.random.randint(0, 50, 64)
        df.to_parquet(str(tmp_path / f"part-{part}.parquet"))
    r = subprocess.run(
        [sys.executable, "-m", "deeprec_amd.models.runner", "--model",
         "dlrm", "--steps", "100", "--batch_size", "32", "--no_bf16",
         "--ev", "--emb_fusion", "--op_fusion", "--group_embedding",
         "--adaptive_emb",
         "--workqueue", str(tmp_path / "part-*.parquet")],
        capture_output=True, text=True, timeout=240)
    # 4 batches exist; --steps 100 must stop at input exhaustion, not die
    assert r.returncode == 0 and "RESULT" in r.stdout, r.stderr[-800:]


def test_runner_distributed_gloo_two_ranks():
    """The runner's world>1 path (init_distributed, sharded collection,
    dense reducer, rank-0 RESULT) through a real 2-process torchrun
    launch on gloo — the CPU analog of the driver's multi-GPU launch."""
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29573", "-m", "deeprec_amd.models.runner",
         "--model", "dlrm", "--steps", "3", "--batch_size", "64",
         "--no_bf16", "--sharded", "--log_steps", "1"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout[-400:], r.stderr[-800:])
    assert "RESULT" in r.stdout and "world=2" in r.stdout


def test_runner_eval_loop(capsys):
    """--eval_steps: post-training eval (no inserts) printing
    loss/accuracy/AUC — the reference train.py eval loop."""
    main(["--model", "wdl", "--steps", "3", "--batch_size", "64",
          "--no_bf16", "--eval_steps", "3"])
    out = capsys.readouterr().out
    assert "EVAL model=wdl" in out and "auc=" in out
    auc = float(out.split("auc=")[1].split()[0])
    assert 0.0 <= auc <= 1.0


def test_runner_eval_multitask(capsys):
    """Multi-task models evaluate with the full head structure passed
    to loss_fn (AUC/accuracy on the primary head)."""
    main(["--model", "esmm", "--steps", "2", "--batch_size", "64",
          "--no_bf16", "--eval_steps", "2"])
    out = capsys.readouterr().out
    assert "EVAL model=esmm" in out and "auc=" in out

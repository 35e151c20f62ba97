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

from deeprec_amd.checkpoint.saver import (  # noqa: F401
    Saver,
    latest_checkpoint,
)

from deeprec_amd.training.estimator import Estimator, RunConfig  # noqa: F401
from deeprec_amd.training.graph_step import GraphedTrainStep  # noqa: F401
from deeprec_amd.training.metrics import (  # noqa: F401
    StreamingAccuracy,
    StreamingAUC,
)
from deeprec_amd.training.session import (  # noqa: F401
    CheckpointSaverHook,
    LoggingTensorHook,
    MemoryStatsHook,
    MonitoredTrainingSession,
    ProfilerHook,
    RebalanceHook,
    SessionRunHook,
    StepCounterHook,
    StopAtStepHook,
)

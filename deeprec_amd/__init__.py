"""deeprec_amd — MI355X-native sparse recommendation training framework.

A from-scratch framework with the capability surface of DeepRec
(dynamic-shape hash EmbeddingVariable with feature admission/eviction,
fused sparse ops, incremental checkpoint, collective multi-GPU training,
model zoo) designed MI355X-first: PyTorch-ROCm front-end, hand-written
HIP/CDNA4 kernels for the sparse hot path, RCCL over xGMI for collectives.

Reference capability map: see SURVEY.md (reference: DeepRec-AI/DeepRec).
"""

__version__ = "0.2.0"

from deeprec_amd.embedding.options import (  # noqa: F401
    EmbeddingVariableOption,
    StorageOption,
    StorageType,
    InitializerOption,
    CounterFilter,
    CBFFilter,
    GlobalStepEvict,
    L2WeightEvict,
)
from deeprec_amd.embedding.variable import (  # noqa: F401
    EmbeddingVariable,
    get_embedding_variable,
)
from deeprec_amd.embedding.lookup import (  # noqa: F401
    embedding_lookup,
    embedding_lookup_sparse,
    safe_embedding_lookup_sparse,
    group_embedding_lookup_sparse,
)
from deeprec_amd.embedding.ragged import RaggedIds  # noqa: F401
from deeprec_amd.embedding.collection import (  # noqa: F401
    EmbeddingCollection,
)
from deeprec_amd.embedding.hash_table import (  # noqa: F401
    BloomFilterAdmitStrategy,
    DistributedHashTable,
    HashTable,
)
from deeprec_amd.embedding.options import CacheStrategy  # noqa: F401
from deeprec_amd.ops.fused_norm import (  # noqa: F401
    fused_l2_normalize,
    fused_layer_norm,
)

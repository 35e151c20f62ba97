from deeprec_amd.embedding.options import *  # noqa: F401,F403
from deeprec_amd.embedding.ragged import RaggedIds  # noqa: F401
from deeprec_amd.embedding.variable import (  # noqa: F401
    EmbeddingVariable, get_embedding_variable, all_embedding_variables,
    reset_registry, GLOBAL_STEP, get_global_step)
from deeprec_amd.embedding.lookup import (  # noqa: F401
    embedding_lookup, embedding_lookup_sparse,
    safe_embedding_lookup_sparse, group_embedding_lookup_sparse)
from deeprec_amd.embedding.hash_table import (  # noqa: F401,E402
    BloomFilterAdmitStrategy, CounterAdmitStrategy,
    DistributedHashTable, HashTable)

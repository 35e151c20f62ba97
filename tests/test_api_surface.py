"""The public API surface promised by docs/MIGRATION.md must resolve —
every symbol a migrating DeepRec user is told to use. Keeps the
migration doc an enforced contract, not prose."""
import importlib

import pytest

SYMBOLS = [
    # embedding variables
    "deeprec_amd:get_embedding_variable",
    "deeprec_amd:EmbeddingVariableOption",
    "deeprec_amd.embedding:EmbeddingVariable",
    "deeprec_amd.embedding.options:CounterFilter",
    "deeprec_amd.embedding.options:CBFFilter",
    "deeprec_amd.embedding.options:GlobalStepEvict",
    "deeprec_amd.embedding.options:L2WeightEvict",
    "deeprec_amd.embedding.options:StorageOption",
    "deeprec_amd.embedding.options:StorageType",
    "deeprec_amd.embedding.options:InitializerOption",
    "deeprec_amd.embedding.options:CacheStrategy",
    "deeprec_amd.embedding.maintenance:EvictionManager",
    "deeprec_amd.embedding.maintenance:MaintenanceHook",
    "deeprec_amd.embedding.cache:BatchCache",
    "deeprec_amd.embedding.cache:LRUCache",
    "deeprec_amd.embedding.cache:LFUCache",
    "deeprec_amd.embedding.cache:make_cache",
    "deeprec_amd.embedding.ssd_kv:SsdKv",
    "deeprec_amd.embedding.extras:MultiHashVariable",
    "deeprec_amd.embedding.extras:DynamicEmbeddingVariable",
    "deeprec_amd.embedding.extras:adaptive_embedding_lookup_sparse",
    # lookups / feature columns
    "deeprec_amd.embedding:embedding_lookup",
    "deeprec_amd.embedding:embedding_lookup_sparse",
    "deeprec_amd.embedding:RaggedIds",
    "deeprec_amd.embedding.lookup:safe_embedding_lookup_sparse",
    "deeprec_amd.embedding.lookup:group_embedding_lookup_sparse",
    "deeprec_amd.feature_column:categorical_column_with_embedding",
    "deeprec_amd.feature_column:categorical_column_with_adaptive_embedding",
    "deeprec_amd.feature_column:sequence_categorical_column_with_embedding",
    "deeprec_amd.feature_column:group_embedding_column_scope",
    "deeprec_amd.feature_column:bucketized_column",
    "deeprec_amd.feature_column:crossed_column",
    "deeprec_amd.feature_column:weighted_categorical_column",
    "deeprec_amd.feature_column:indicator_column",
    "deeprec_amd.feature_column:categorical_column_with_identity",
    "deeprec_amd.feature_column:categorical_column_with_vocabulary_list",
    "deeprec_amd.feature_column:shared_embedding_columns",
    "deeprec_amd.feature_column:InputLayer",
    "deeprec_amd.embedding.hash_table:HashTable",
    "deeprec_amd.embedding.hash_table:DistributedHashTable",
    # training
    "deeprec_amd.optimizers:AdamAsyncOptimizer",
    "deeprec_amd.optimizers:AdagradDecayOptimizer",
    "deeprec_amd.optimizers:AdamWOptimizer",
    "deeprec_amd.optimizers:FtrlOptimizer",
    "deeprec_amd.optimizers:make_optimizer",
    "deeprec_amd.training:MonitoredTrainingSession",
    "deeprec_amd.training:Estimator",
    "deeprec_amd.training.session:ProfilerHook",
    "deeprec_amd.training.session:StepCounterHook",
    "deeprec_amd.training.session:LoggingTensorHook",
    "deeprec_amd.training.graph_step:GraphedTrainStep",
    "deeprec_amd.training.schedules:exponential_decay",
    "deeprec_amd.training.schedules:polynomial_decay",
    "deeprec_amd.training.schedules:piecewise_constant",
    "deeprec_amd.training.schedules:LearningRateScheduleHook",
    "deeprec_amd.training.metrics:StreamingAUC",
    "deeprec_amd.training.cluster:parse_tf_config",
    "deeprec_amd.training.cluster:start_ps",
    "deeprec_amd.training.cluster:worker_embeddings",
    "deeprec_amd.checkpoint.saver:Saver",
    "deeprec_amd.checkpoint.saver:latest_checkpoint",
    # distributed
    "deeprec_amd.parallel:DenseGradAllreducer",
    "deeprec_amd.parallel:ShardedEmbeddingCollection",
    "deeprec_amd.parallel:init_distributed",
    "deeprec_amd.parallel.ps:PsServer",
    "deeprec_amd.parallel.ps:PsClient",
    "deeprec_amd.parallel.ps:PsShardedEmbedding",
    "deeprec_amd.parallel.elastic:live_resize",
    "deeprec_amd.parallel.hierarchical:hierarchical_all_to_all",
    # data
    "deeprec_amd.data.parquet:ParquetDataset",
    "deeprec_amd.data.parquet:CsvDataset",
    "deeprec_amd.data.parquet:WorkQueue",
    "deeprec_amd.data.kafka:KafkaDataset",
    "deeprec_amd.data.kafka_wire:MiniKafkaBroker",
    "deeprec_amd.data.prefetch:PrefetchIterator",
    "deeprec_amd.data.compression:compressed_forward",
    # serving
    "deeprec_amd.serving.predictor:Predictor",
    "deeprec_amd.serving.predictor:SessionGroup",
    "deeprec_amd.serving.server:create_app",
    "deeprec_amd.serving.server:DynamicBatcher",
    "deeprec_amd.serving.redis_store:RedisFeatureStore",
    "deeprec_amd.serving.redis_store:MiniRedisServer",
    "deeprec_amd.serving.remote_kv:attach_remote_store",
    "deeprec_amd.serving.device_placement:move_embeddings_to_cpu",
    # fp8 / quantization
    "deeprec_amd.ops.fp8:Fp8Linear",
    "deeprec_amd.ops.fp8:convert_mlp_to_fp8",
    "deeprec_amd.ops.fp8:quantize_fp8_rows",
    "deeprec_amd.ops.fp8:Fp8MlpConverter",
    # models
    "deeprec_amd.models:MODEL_REGISTRY",
]


@pytest.mark.parametrize("spec", SYMBOLS)
def test_symbol_resolves(spec):
    mod, attr = spec.split(":")
    m = importlib.import_module(mod)
    assert hasattr(m, attr), spec


def test_model_registry_complete():
    from deeprec_amd.models import MODEL_REGISTRY
    expected = {"dlrm", "wdl", "deepfm", "dcn", "dcnv2", "mlperf_dlrm_dcn",
                "dssm", "mmoe", "esmm", "dbmtl", "ple", "masknet",
                "simple_multitask", "din", "dien", "bst"}
    assert expected <= set(MODEL_REGISTRY), \
        expected - set(MODEL_REGISTRY)

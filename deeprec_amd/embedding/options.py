"""EmbeddingVariable option classes.

Capability parity with the reference's Python option surface
(reference: python/ops/variables.py:180-304 — EmbeddingVariableOption,
StorageOption, InitializerOption, GlobalStepEvict, L2WeightEvict,
CounterFilter, CBFFilter), re-designed for a single-process-per-GPU
PyTorch-ROCm runtime.
"""
from __future__ import annotations

import enum
from dataclasses import dataclass, field
from typing import Optional


class StorageType(enum.Enum):
    """Where an EmbeddingVariable's key/value store lives.

    Reference enum: core/framework/embedding/config.proto:5-27. The MI355X
    design collapses the tier zoo to the tiers that exist on this platform:
    HBM (288 GB per GPU), DRAM (host), HBM_DRAM (hot HBM tier + host cold
    tier staged over pinned hipMemcpyAsync).
    """

    DEFAULT = 0      # pick by device: HBM on GPU, DRAM on CPU
    DRAM = 1
    HBM = 2
    HBM_DRAM = 3
    # hot HBM + pinned-DRAM middle tier + append-only SSD files with
    # compaction (reference: HbmDramSsdStorage, hbm_dram_ssd_storage.h);
    # storage_size = [hbm_bytes, dram_bytes], storage_path = SSD dir
    HBM_DRAM_SSD = 4


class CacheStrategy(enum.Enum):
    """Hot-key tracking for multi-tier placement (reference: config.proto:40-43)."""

    LRU = 1
    LFU = 2


@dataclass
class StorageOption:
    storage_type: StorageType = StorageType.DEFAULT
    storage_path: Optional[str] = None
    # per-tier byte budgets; for HBM_DRAM, storage_size[0] caps the HBM hot tier
    storage_size: Optional[list] = None
    cache_strategy: CacheStrategy = CacheStrategy.LFU


@dataclass
class InitializerOption:
    """EV default-value initializer.

    The reference materializes a [default_value_dim, dim] matrix and a new
    key k is initialized from row (k % default_value_dim)
    (reference: embedding_var.h default_value_ usage). Same contract here.
    """

    initializer: Optional[object] = None  # callable(tensor) -> None, in-place
    default_value_dim: int = 4096
    default_value_no_permission: float = 0.0


@dataclass
class CounterFilter:
    """Admit a key into the table only after `filter_freq` occurrences.

    Reference: counter_filter_policy.h:25, docs Feature-Filter.md.
    """

    filter_freq: int = 0


@dataclass
class CBFFilter:
    """Counting-bloom-filter admission (probabilistic, memory-bounded).

    Reference: bloom_filter_policy.h:33. Pre-admission counts live in a
    counting bloom filter instead of per-key metadata.
    """

    filter_freq: int = 0
    max_element_size: int = 0
    false_positive_probability: float = 0.01
    counter_type: str = "uint16"


@dataclass
class GlobalStepEvict:
    """Evict keys whose last-update step is older than steps_to_live.

    Reference: globalstep_shrink_policy.h; shrink runs at checkpoint save.
    """

    steps_to_live: int = 0


@dataclass
class L2WeightEvict:
    """Evict keys whose value L2-norm is below the threshold.

    Reference: l2weight_shrink_policy.h.
    """

    l2_weight_threshold: float = -1.0


@dataclass
class EmbeddingVariableOption:
    """Aggregate EV configuration (reference: EmbeddingVariableOption,
    python/ops/variables.py:281-304 + EmbeddingConfig embedding_config.h)."""

    filter_option: Optional[object] = None   # CounterFilter | CBFFilter | None
    evict_option: Optional[object] = None    # GlobalStepEvict | L2WeightEvict | None
    storage_option: StorageOption = field(default_factory=StorageOption)
    init_option: InitializerOption = field(default_factory=InitializerOption)
    # initial hash-table capacity (entries); grows automatically
    init_capacity: int = 1 << 17
    # API compatibility: this engine ALWAYS records per-key freq/version
    # (SoA int32/int64 arrays are cheap in HBM and feed admission,
    # eviction and LFU/LRU tier placement); the reference flags are
    # accepted so option structs port unchanged
    record_freq: bool = False
    record_version: bool = False

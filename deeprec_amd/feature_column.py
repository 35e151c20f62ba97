"""feature_column API subset.

Capability parity with the reference's feature-column surface used by the
modelzoo (reference: python/feature_column/feature_column_v2.py —
numeric_column, categorical_column_with_embedding:2080,
categorical_column_with_hash_bucket, embedding_column,
shared_embedding_columns, sequence_categorical_column_with_embedding,
group_embedding_column_scope:4239, and tf.feature_column.input_layer).

Design: columns are declarative specs; `input_layer(features, columns)`
builds/looks-up the backing EmbeddingVariables. Embedding columns declared
inside a `group_embedding_column_scope` share one EmbeddingCollection (the
fused multi-table path).
"""
from __future__ import annotations

import contextlib
import threading
from dataclasses import dataclass
from typing import List, Optional, Sequence

import torch

from deeprec_amd.embedding.collection import EmbeddingCollection
from deeprec_amd.embedding.options import EmbeddingVariableOption
from deeprec_amd.embedding.ragged import RaggedIds
from deeprec_amd.embedding.variable import get_embedding_variable
from deeprec_amd.embedding.lookup import (
    embedding_lookup, embedding_lookup_sparse)

_SCOPE = threading.local()


@dataclass
class NumericColumn:
    key: str
    shape: int = 1
    normalizer_fn: Optional[object] = None


@dataclass
class CategoricalColumn:
    key: str
    hash_bucket_size: Optional[int] = None   # None => EV (unbounded keys)
    ev_option: Optional[EmbeddingVariableOption] = None
    sequence: bool = False


@dataclass
class EmbeddingColumnSpec:
    categorical: CategoricalColumn
    dimension: int
    combiner: str = "mean"
    shared_name: Optional[str] = None
    group: Optional[str] = None


@dataclass
class BucketizedColumn:
    source: NumericColumn
    boundaries: Sequence


@dataclass
class CrossedColumn:
    keys: Sequence
    hash_bucket_size: int
    # behaves as a categorical for embedding_column()/indicator_column()
    ev_option: Optional[EmbeddingVariableOption] = None
    sequence: bool = False

    @property
    def key(self):
        return "_X_".join(k if isinstance(k, str) else k.key
                          for k in self.keys)


@dataclass
class WeightedCategoricalColumn:
    categorical: CategoricalColumn
    weight_feature_key: str

    @property
    def key(self):
        return self.categorical.key

    @property
    def hash_bucket_size(self):
        return self.categorical.hash_bucket_size

    @property
    def ev_option(self):
        return self.categorical.ev_option

    @property
    def sequence(self):
        return self.categorical.sequence


@dataclass
class IndicatorColumn:
    categorical: CategoricalColumn


@dataclass
class AdaptiveEmbeddingColumnSpec:
    """Hot ids in an EV, cold ids in a fixed hash bucket (reference:
    categorical_column_with_adaptive_embedding, feature_column_v2.py:2089
    + adaptive_embedding_lookup_sparse, embedding_ops.py:668)."""
    categorical: CategoricalColumn
    dimension: int
    static_bucket_size: int
    combiner: str = "mean"


def numeric_column(key: str, shape: int = 1, normalizer_fn=None):
    return NumericColumn(key, shape, normalizer_fn)


def bucketized_column(source: NumericColumn, boundaries):
    """Numeric -> ordinal bucket categorical (stock-TF surface used
    throughout the reference modelzoo)."""
    return BucketizedColumn(source, list(boundaries))


def crossed_column(keys, hash_bucket_size: int):
    """Feature cross hashed into a fixed bucket space (stock-TF surface;
    the reference WDL wide side is built from crosses)."""
    return CrossedColumn(list(keys), hash_bucket_size)


def weighted_categorical_column(categorical: CategoricalColumn,
                                weight_feature_key: str):
    return WeightedCategoricalColumn(categorical, weight_feature_key)


def indicator_column(categorical: CategoricalColumn):
    """Multi-hot dense encoding (requires hash_bucket_size — an
    unbounded EV id space has no fixed one-hot width)."""
    assert categorical.hash_bucket_size, \
        "indicator_column needs a bounded (hash-bucket) id space"
    return IndicatorColumn(categorical)


def categorical_column_with_identity(key: str, num_buckets: int):
    return CategoricalColumn(key, num_buckets)


def categorical_column_with_vocabulary_list(key: str, vocabulary_list):
    col = CategoricalColumn(key, len(vocabulary_list))
    col.vocabulary = {v: i for i, v in enumerate(vocabulary_list)}
    return col


def categorical_column_with_adaptive_embedding(key: str,
                                               hash_bucket_size: int,
                                               dtype=torch.int64,
                                               ev_option=None):
    col = CategoricalColumn(key, None, ev_option)
    col.adaptive_bucket = hash_bucket_size
    return col


def adaptive_embedding_column(categorical: CategoricalColumn,
                              dimension: int, combiner: str = "mean"):
    bucket = getattr(categorical, "adaptive_bucket", None)
    assert bucket, ("use categorical_column_with_adaptive_embedding for "
                    "adaptive columns")
    return AdaptiveEmbeddingColumnSpec(categorical, dimension, bucket,
                                       combiner)


def categorical_column_with_embedding(key: str, dtype=torch.int64,
                                      ev_option=None):
    """EV-backed categorical column (unbounded id space) — the reference's
    signature addition over stock TF."""
    return CategoricalColumn(key, None, ev_option)


def categorical_column_with_hash_bucket(key: str, hash_bucket_size: int):
    return CategoricalColumn(key, hash_bucket_size)


def sequence_categorical_column_with_embedding(key: str, dtype=torch.int64,
                                               ev_option=None):
    return CategoricalColumn(key, None, ev_option, sequence=True)


def embedding_column(categorical: CategoricalColumn, dimension: int,
                     combiner: str = "mean"):
    group = getattr(_SCOPE, "group", None)
    return EmbeddingColumnSpec(categorical, dimension, combiner,
                               group=group)


def shared_embedding_columns(categoricals: Sequence[CategoricalColumn],
                             dimension: int, combiner: str = "mean",
                             shared_embedding_collection_name: str = None):
    name = shared_embedding_collection_name or "_".join(
        c.key for c in categoricals) + "_shared"
    return [EmbeddingColumnSpec(c, dimension, combiner, shared_name=name)
            for c in categoricals]


@contextlib.contextmanager
def group_embedding_column_scope(name: str = "group"):
    """Embedding columns created inside share one EmbeddingCollection
    (reference: group_embedding_column_scope, feature_column_v2.py:4239)."""
    prev = getattr(_SCOPE, "group", None)
    _SCOPE.group = name
    try:
        yield
    finally:
        _SCOPE.group = prev


class InputLayer(torch.nn.Module):
    """Materializes columns into a dense [B, total_dim] tensor.

    features: dict key -> tensor ([B] / [B, k] int64 for categorical,
    float tensor or RaggedIds for numeric/sequence).
    """

    def __init__(self, columns: Sequence, device="cpu", out_dtype=None):
        super().__init__()
        self.columns = list(columns)
        self.device_ = torch.device(device)
        self.out_dtype = out_dtype
        self._collections = {}
        self._evs = {}
        # build grouped collections
        groups = {}
        for col in self.columns:
            if isinstance(col, EmbeddingColumnSpec) and col.group:
                groups.setdefault((col.group, col.dimension), []).append(col)
        for (gname, dim), cols in groups.items():
            self._collections[(gname, dim)] = EmbeddingCollection(
                f"fc/{gname}/{dim}", [c.categorical.key for c in cols], dim,
                ev_option=cols[0].categorical.ev_option,
                combiners=[c.combiner for c in cols], device=self.device_)

    def _ev_for(self, col: EmbeddingColumnSpec):
        name = col.shared_name or f"fc/{col.categorical.key}"
        if name not in self._evs:
            self._evs[name] = get_embedding_variable(
                name, col.dimension, ev_option=col.categorical.ev_option,
                device=self.device_)
        return self._evs[name]

    def embedding_variables(self):
        return list(self._collections.values()) + list(self._evs.values())

    def forward(self, features: dict, train: bool = True) -> torch.Tensor:
        outs: List[torch.Tensor] = []
        group_inputs = {}
        group_cols = {}
        for col in self.columns:
            if isinstance(col, NumericColumn):
                x = features[col.key].float()
                if col.normalizer_fn:
                    x = col.normalizer_fn(x)
                outs.append(x.reshape(x.shape[0], -1))
            elif isinstance(col, BucketizedColumn):
                x = features[col.source.key].float().reshape(-1)
                bnd = torch.as_tensor(col.boundaries, dtype=torch.float32,
                                      device=x.device)
                idx = torch.bucketize(x, bnd)
                outs.append(torch.nn.functional.one_hot(
                    idx, len(col.boundaries) + 1).float())
            elif isinstance(col, IndicatorColumn):
                ids = features[col.categorical.key]
                if ids.dim() == 1:
                    ids = ids.unsqueeze(1)
                bucket = col.categorical.hash_bucket_size
                z = torch.zeros(ids.shape[0], bucket, device=ids.device)
                z.scatter_(1, (ids % bucket).long(), 1.0)
                outs.append(z)
            elif isinstance(col, AdaptiveEmbeddingColumnSpec):
                outs.append(self._adaptive(col, features, train))
            elif isinstance(col, EmbeddingColumnSpec):
                key = col.categorical.key
                if isinstance(col.categorical, CrossedColumn):
                    ids = self._crossed_ids(col.categorical, features)
                else:
                    ids = features[key]
                if col.group:
                    group_inputs.setdefault((col.group, col.dimension),
                                            []).append(self._as_ragged(
                                                ids, col))
                    group_cols.setdefault((col.group, col.dimension),
                                          []).append(col)
                    outs.append(None)  # placeholder, filled below
                else:
                    outs.append(self._single(col, ids, train, features))
        # grouped lookups (one fused call per collection)
        fills = {}
        for gk, sp_list in group_inputs.items():
            coll = self._collections[gk]
            cat = coll.lookup(sp_list, out_dtype=self.out_dtype, train=train)
            dim = gk[1]
            for i, col in enumerate(group_cols[gk]):
                fills[id(col)] = cat[:, i * dim:(i + 1) * dim]
        gi = 0
        for i, col in enumerate(self.columns):
            if outs[i] is None:
                outs[i] = fills[id(col)]
                gi += 1
        ref = next(o for o in outs if o is not None)
        return torch.cat([o.to(ref.dtype) for o in outs], dim=1)

    def _crossed_ids(self, cross: CrossedColumn, features):
        h = None
        for k in cross.keys:
            key = k if isinstance(k, str) else k.key
            v = features[key]
            if v.dim() > 1:
                v = v.reshape(v.shape[0], -1)[:, 0]
            v = v.long()
            h = v if h is None else h * 1000003 + v
        return (h % cross.hash_bucket_size).unsqueeze(1)

    def _adaptive(self, col, features, train):
        from deeprec_amd.embedding.extras import (
            adaptive_embedding_lookup_sparse)
        key = col.categorical.key
        name = f"fc/adaptive/{key}"
        if name not in self._evs:
            self._evs[name] = get_embedding_variable(
                name, col.dimension, ev_option=col.categorical.ev_option,
                device=self.device_)
        static_name = f"_adaptive_static_{key}"
        if not hasattr(self, static_name):
            w = torch.nn.Parameter(torch.randn(
                col.static_bucket_size, col.dimension,
                device=self.device_) / (col.dimension ** 0.5))
            setattr(self, static_name, w)
        sp = self._as_ragged(features[key], None)
        return adaptive_embedding_lookup_sparse(
            self._evs[name], getattr(self, static_name), sp,
            threshold=getattr(col.categorical, "adaptive_threshold", 3),
            combiner=col.combiner, train=train)

    def _as_ragged(self, ids, col) -> RaggedIds:
        weights = None
        if col is not None and isinstance(col, EmbeddingColumnSpec) and \
                isinstance(col.categorical, WeightedCategoricalColumn):
            weights = self._pending_weights
        if isinstance(ids, RaggedIds):
            return ids
        if ids.dim() == 1:
            ids = ids.unsqueeze(1)
        bucket = None if col is None else \
            getattr(col.categorical, "hash_bucket_size", None)
        if bucket:
            ids = ids % bucket
        sp = RaggedIds.from_dense(ids)
        if weights is not None:
            return RaggedIds(sp.values, sp.offsets,
                             weights.reshape(-1).float())
        return sp

    def _single(self, col, ids, train, features=None):
        ev = self._ev_for(col)
        if col.categorical.sequence:
            return embedding_lookup(ev, ids, train=train).flatten(1)
        self._pending_weights = None
        if isinstance(col.categorical, WeightedCategoricalColumn) and \
                features is not None:
            self._pending_weights = features[
                col.categorical.weight_feature_key]
        sp = self._as_ragged(ids, col)
        self._pending_weights = None
        return embedding_lookup_sparse(ev, sp, combiner=col.combiner,
                                       train=train)


def input_layer(features: dict, columns: Sequence, device="cpu",
                train: bool = True) -> torch.Tensor:
    """One-shot functional form (builds EVs on first use via the registry)."""
    layer = InputLayer(columns, device)
    return layer(features, train)

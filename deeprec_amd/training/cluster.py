"""TF_CONFIG-style cluster roles for multi-node async-PS training.

Capability parity with the reference's distributed Estimator story
(modelzoo/dlrm/train.py:858-913: TF_CONFIG -> ClusterSpec +
replica_device_setter): a job is {"cluster": {"ps": [...], "worker":
[...]}, "task": {"type": ..., "index": ...}}; PS tasks host EV shards
behind the pull/push plane (parallel/ps.py), workers train with
PsShardedEmbedding for sparse parameters and (optionally) a
torch.distributed group for dense gradients.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional, Tuple

from deeprec_amd.parallel.ps import PsClient, PsServer, PsShardedEmbedding


def parse_tf_config(cfg: Optional[str] = None) -> dict:
    cfg = cfg if cfg is not None else os.environ.get("TF_CONFIG", "{}")
    d = json.loads(cfg) if isinstance(cfg, str) else dict(cfg)
    cluster = d.get("cluster", {})
    task = d.get("task", {"type": "worker", "index": 0})
    return {"ps": list(cluster.get("ps", [])),
            "worker": list(cluster.get("worker", [])),
            "type": task.get("type", "worker"),
            "index": int(task.get("index", 0))}


def _split_addr(addr: str) -> Tuple[str, int]:
    host, port = addr.rsplit(":", 1)
    return host, int(port)


def start_ps(config: dict, tables: Dict[str, int],
             optimizer: str = "adagrad", lr: float = 0.1,
             checkpoint_dir: Optional[str] = None) -> PsServer:
    """Run THIS task's PS role (returns the live server; call
    .close() or serve until the chief signals shutdown)."""
    host, port = _split_addr(config["ps"][config["index"]])
    return PsServer(tables, ps_index=config["index"],
                    optimizer=optimizer, lr=lr, host=host, port=port,
                    checkpoint_dir=checkpoint_dir)


def worker_embeddings(config: dict,
                      tables: Dict[str, int]) -> Dict[str,
                                                      PsShardedEmbedding]:
    """Worker-side sparse parameter facades, one per table, routed over
    every PS task."""
    client = PsClient([_split_addr(a) for a in config["ps"]])
    return {name: PsShardedEmbedding(client, name, dim)
            for name, dim in tables.items()}

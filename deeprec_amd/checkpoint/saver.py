"""Checkpointing: full + incremental save/restore.

Capability parity with the reference:
- full checkpoint: dense weights + optimizer state + global step + every
  EV/collection exported as <name>-keys/-values/-freqs/-versions tensors
  plus the slot-aligned optimizer slabs; keys are additionally tagged with
  a bucket id (key % NUM_BUCKETS, reference kSavedPartitionNum=1000) so a
  restore can repartition to ANY shard count (each rank filters the keys
  it owns on load — reference: embedding_var_ckpt_data.h:57).
- incremental checkpoint: after a full save the EVs record every touched
  id (reference: RecordSparseIndices, incr_save_restore_ops.cc:22);
  incremental_save dumps only those keys' current rows; restore = full
  checkpoint + ordered replay of deltas (reference: IncrSave/IncrRestore,
  python/training/incremental_saver.py:78-127).

Format: a directory per checkpoint; tensors in safetensors files, small
metadata in JSON. Multi-rank training writes per-rank EV shard files; the
dense state is written by rank 0 only (ranks hold replicas).
"""
from __future__ import annotations

import glob
import json
import os
import shutil
import time
from typing import Dict, List, Optional

import torch
from safetensors.torch import load_file, save_file

NUM_BUCKETS = 1000  # reference kSavedPartitionNum

# optimizer slab names we persist when present
_SLAB_NAMES = ["adagrad_accum", "adagrad_decay_period", "adam_m", "adam_v",
               "ftrl_accum", "ftrl_linear"]


def _is_collection(ev) -> bool:
    return hasattr(ev, "export_tables")


def _ev_like_name(ev) -> str:
    return ev.name.replace("/", "__")


class Saver:
    """Full-checkpoint saver/restorer.

    `embedding_variables` may contain EmbeddingVariable, EmbeddingCollection
    or their sharded wrappers; `module` is the dense nn.Module; `optimizer`
    our Optimizer wrapper (dense state + step counters).
    """

    def __init__(self, module: Optional[torch.nn.Module] = None,
                 embedding_variables: Optional[List] = None,
                 optimizer=None, keep_checkpoint_max: int = 5,
                 rank: int = 0, world_size: int = 1,
                 save_filtered: Optional[bool] = None):
        self.module = module
        self.evs = list(embedding_variables or [])
        self.optimizer = optimizer
        self.keep_checkpoint_max = keep_checkpoint_max
        self.rank = rank
        self.world_size = world_size
        # save sub-threshold admission counters so a restore continues
        # the filter exactly (reference: TF_EV_SAVE_FILTERED_FEATURES,
        # embedding_var.h:533)
        if save_filtered is None:
            save_filtered = os.environ.get(
                "DEEPREC_EV_SAVE_FILTERED_FEATURES",
                "0") not in ("0", "", "false")
        self.save_filtered = bool(save_filtered)

    # ------------- save -------------
    def save(self, directory: str, global_step: int) -> str:
        path = os.path.join(directory, f"ckpt-{global_step}")
        os.makedirs(path, exist_ok=True)
        if self.rank == 0:
            self._save_dense(path, global_step)
        for ev in self.evs:
            self._save_ev(path, ev)
        if self.rank == 0:
            with open(os.path.join(path, "checkpoint.json"), "w") as f:
                json.dump({"global_step": global_step,
                           "world_size": self.world_size,
                           "num_buckets": NUM_BUCKETS,
                           "timestamp": time.time()}, f)
        # arm incremental recording from this point
        for ev in self.evs:
            base = getattr(ev, "local", ev)
            base.start_sparse_recording()
            base.consume_recorded_ids()
        if self.rank == 0:
            self._cleanup(directory)
        return path

    def _save_dense(self, path: str, global_step: int):
        state = {"global_step": torch.tensor(global_step)}
        if self.module is not None:
            for k, v in self.module.state_dict().items():
                state[f"module/{k}"] = v.detach().cpu().contiguous()
        save_file(state, os.path.join(path, "dense.safetensors"))
        if self.optimizer is not None:
            torch.save(self.optimizer.state_dict(),
                       os.path.join(path, "optimizer.pt"))

    def _ev_payload(self, base) -> Dict[str, torch.Tensor]:
        if self.save_filtered:
            keys, values, freqs, versions, fkeys, ffreqs = \
                base.export(include_filtered=True)
        else:
            keys, values, freqs, versions = \
                base.export(include_filtered=False)
            fkeys = None
        present = [n for n in _SLAB_NAMES if n in base.storage.slabs]
        slab_rows = base.storage.export_slabs(present) if present else []
        payload = {
            "keys": keys.cpu(),
            "values": values.cpu().float(),
            "freqs": freqs.cpu().to(torch.int64),
            "versions": versions.cpu(),
            "buckets": (keys.cpu() % NUM_BUCKETS).to(torch.int32),
        }
        if fkeys is not None:
            payload["filtered_keys"] = fkeys.cpu()
            payload["filtered_freqs"] = ffreqs.cpu().to(torch.int64)
            cbf = getattr(base.storage, "_cbf", None)
            if cbf is not None:
                # counting-bloom pre-admission state is the filter itself
                import numpy as _np
                payload["cbf_counters"] = torch.from_numpy(
                    cbf.counters.astype(_np.int64))
        for n, rows in zip(present, slab_rows):
            payload[f"slab/{n}"] = rows.cpu()
            # slab fill value for keys admitted AFTER restore (e.g.
            # adagrad's initial_accumulator) — without it, post-restore
            # admissions start from 0.0 and trajectories diverge
            payload[f"slabinit/{n}"] = torch.tensor(
                float(base.storage._slab_init.get(n, 0.0)))
        return payload

    def _save_ev(self, path: str, ev):
        base = getattr(ev, "local", ev)  # sharded wrappers expose .local
        name = _ev_like_name(base if not hasattr(ev, "local") else ev)
        fn = os.path.join(path, f"ev-{name}-part{self.rank}.safetensors")
        save_file(self._ev_payload(base), fn)

    def _cleanup(self, directory: str):
        def step_of(p):
            tail = p.rsplit("-", 1)[1].replace(".incr", "")
            return int(tail) if tail.isdigit() else -1

        cks = sorted((c for c in glob.glob(os.path.join(directory,
                                                        "ckpt-*"))
                      if not c.endswith(".incr")), key=step_of)
        while len(cks) > self.keep_checkpoint_max:
            shutil.rmtree(cks.pop(0), ignore_errors=True)
        if cks:
            # incremental deltas older than the oldest kept full ckpt
            # can never be replayed again (restore = full + NEWER incr)
            oldest = step_of(cks[0])
            for inc in glob.glob(os.path.join(directory, "ckpt-*.incr")):
                if step_of(inc) < oldest:
                    shutil.rmtree(inc, ignore_errors=True)

    # ------------- restore -------------
    def restore(self, ckpt_path: str) -> int:
        meta = json.load(open(os.path.join(ckpt_path, "checkpoint.json")))
        if self.module is not None:
            state = load_file(os.path.join(ckpt_path, "dense.safetensors"))
            module_state = {k[len("module/"):]: v for k, v in state.items()
                            if k.startswith("module/")}
            self.module.load_state_dict(module_state)
        opt_file = os.path.join(ckpt_path, "optimizer.pt")
        if self.optimizer is not None and os.path.exists(opt_file):
            self.optimizer.load_state_dict(
                torch.load(opt_file, weights_only=False))
        for ev in self.evs:
            self._restore_ev(ckpt_path, ev)
        # replay incremental deltas saved after this full checkpoint
        step = meta["global_step"]
        base_dir = os.path.dirname(ckpt_path)
        incrs = sorted(
            (p for p in glob.glob(os.path.join(base_dir, "ckpt-*.incr"))
             if int(os.path.basename(p)[5:-5]) > step),
            key=lambda p: int(os.path.basename(p)[5:-5]))
        for incr in incrs:
            self._restore_ev_files(incr, replay=True)
            step = int(os.path.basename(incr)[5:-5])
        from deeprec_amd.embedding.variable import GLOBAL_STEP
        GLOBAL_STEP.value = step
        return step

    def _owned_mask(self, ev, keys: torch.Tensor) -> torch.Tensor:
        if hasattr(ev, "world") and ev.world > 1:
            if _is_collection(ev):
                # collections tag keys with table_id << KEY_BITS and route
                # on the raw id — strip the tag before the mod
                from deeprec_amd.embedding.collection import KEY_BITS
                raw = keys & ((1 << KEY_BITS) - 1)
                return (raw % ev.world) == ev.rank
            # plain sharded EVs route lookups by the FULL key % world
            # (sharded_embedding.py _ShardedPooledLookup); masking here
            # would strand ids >= 2^48 or negative ids (64-bit feature
            # hashing) on a rank that lookups never probe
            return (keys % ev.world) == ev.rank
        return torch.ones(keys.numel(), dtype=torch.bool)

    def _restore_ev(self, ckpt_path: str, ev):
        name = _ev_like_name(ev)
        files = sorted(glob.glob(
            os.path.join(ckpt_path, f"ev-{name}-part*.safetensors")))
        if not files:
            raise FileNotFoundError(f"no EV shard files for {ev.name} "
                                    f"in {ckpt_path}")
        base = getattr(ev, "local", ev)
        for fn in files:
            data = load_file(fn)
            keys = data["keys"]
            mask = self._owned_mask(ev, keys)
            slab_rows = {k[len("slab/"):]: v[mask] for k, v in data.items()
                         if k.startswith("slab/")}
            self._precreate_slabs(base, data, slab_rows)
            base.storage.import_(
                keys[mask].to(base.device),
                data["values"][mask].to(base.device),
                data["freqs"][mask], data["versions"][mask],
                slab_rows=slab_rows or None)
            self._restore_filtered(base, ev, data)

    def _precreate_slabs(self, base, data, slab_rows):
        for n, rows in slab_rows.items():
            init = data.get(f"slabinit/{n}")
            base.storage.get_slab(n, rows.shape[1],
                                  float(init) if init is not None else 0.0)

    def _restore_filtered(self, base, ev, data):
        cbf_state = data.get("cbf_counters")
        cbf = getattr(base.storage, "_cbf", None)
        if cbf_state is not None and cbf is not None:
            cbf.counters[:] = cbf_state.numpy().astype(
                cbf.counters.dtype)
        fk = data.get("filtered_keys")
        if fk is None or fk.numel() == 0:
            return
        if not hasattr(base.storage, "import_filtered"):
            return
        fmask = self._owned_mask(ev, fk)
        base.storage.import_filtered(fk[fmask].to(base.device),
                                     data["filtered_freqs"][fmask])

    def _restore_ev_files(self, path: str, replay: bool = False):
        for ev in self.evs:
            name = _ev_like_name(ev)
            files = sorted(glob.glob(
                os.path.join(path, f"ev-{name}-part*.safetensors")))
            base = getattr(ev, "local", ev)
            for fn in files:
                data = load_file(fn)
                keys = data["keys"]
                mask = self._owned_mask(ev, keys)
                slab_rows = {k[len("slab/"):]: v[mask]
                             for k, v in data.items()
                             if k.startswith("slab/")}
                self._precreate_slabs(base, data, slab_rows)
                base.storage.import_(
                    keys[mask].to(base.device),
                    data["values"][mask].to(base.device),
                    data["freqs"][mask], data["versions"][mask],
                    slab_rows=slab_rows or None)

    # ------------- incremental -------------
    def incremental_save(self, directory: str, global_step: int) -> str:
        """Dump rows of ids touched since the last (full or incremental)
        save. Requires a prior full save() to arm recording."""
        path = os.path.join(directory, f"ckpt-{global_step}.incr")
        os.makedirs(path, exist_ok=True)
        for ev in self.evs:
            base = getattr(ev, "local", ev)
            ids = base.consume_recorded_ids()
            name = _ev_like_name(ev)
            fn = os.path.join(path,
                              f"ev-{name}-part{self.rank}.safetensors")
            payload = self._ev_rows_payload(base, ids)
            save_file(payload, fn)
        if self.rank == 0:
            with open(os.path.join(path, "checkpoint.json"), "w") as f:
                json.dump({"global_step": global_step,
                           "incremental": True}, f)
        return path

    def _ev_rows_payload(self, base, ids: torch.Tensor):
        ids = ids.to(base.device)
        slots = base.storage.lookup(ids)
        adm = slots >= 0
        ids, slots = ids[adm], slots[adm]
        values = base.storage.gather(ids, slots)
        freqs = base.storage.frequencies(ids)
        versions = base.storage.versions(ids)
        payload = {
            "keys": ids.cpu(), "values": values.cpu().float(),
            "freqs": freqs.cpu().to(torch.int64),
            "versions": versions.cpu(),
            "buckets": (ids.cpu() % NUM_BUCKETS).to(torch.int32),
        }
        sl = slots.long()
        for n in _SLAB_NAMES:
            if n in base.storage.slabs:
                payload[f"slab/{n}"] = base.storage.slabs[n][sl].cpu()
        return payload


def latest_checkpoint(directory: str) -> Optional[str]:
    cks = [p for p in glob.glob(os.path.join(directory, "ckpt-*"))
           if not p.endswith(".incr")
           and os.path.exists(os.path.join(p, "checkpoint.json"))]
    if not cks:
        return None
    return max(cks, key=lambda p: int(os.path.basename(p).split("-")[1]))

"""Checkpoint / model dump + load.

Reference contract: a model dump is the set of every
``(paramId, value)`` pair of each shard, unordered
(``ParameterServerLogic.close`` -> PSOut stream); a load
(``transformWithModelLoad``) feeds such pairs back through the partitioner,
independent of the shard count that wrote them.

Concrete format (one file per shard, little-endian):

    magic   8 bytes  b"FPSCKPT1"
    dim     int64
    dtype   int64    (0 = f32)
    count   int64
    ids     count * int64
    values  count * dim * f32

Unordered on read; re-partitioned on load, so a checkpoint written with N
shards loads onto M shards.  Optimizer state (Adagrad G) dumps as a second
file with the same layout (suffix ``.accum``).
"""
from __future__ import annotations

import json
import os
import struct
from typing import List, Optional, Tuple

import torch

from ..table import ShardedTable

MAGIC = b"FPSCKPT1"
MANIFEST = "MANIFEST.json"


def _write_one(path: str, ids: torch.Tensor, values: torch.Tensor) -> None:
    """Crash-safe single-file write: tmp + fsync + atomic ``os.replace``
    — a dump interrupted mid-write can never leave a truncated ``.fps``
    file under the final name (§5.3's recovery must survive the failure
    it recovers from)."""
    ids = ids.contiguous().cpu()
    values = values.contiguous().cpu().to(torch.float32)
    count, dim = values.shape
    tmp = path + ".tmp"
    with open(tmp, "wb") as f:
        f.write(MAGIC)
        f.write(struct.pack("<qqq", dim, 0, count))
        f.write(ids.numpy().tobytes())
        f.write(values.numpy().tobytes())
        f.flush()
        os.fsync(f.fileno())
    os.replace(tmp, path)


def _write_manifest(directory: str, n_shards: int, dim: int,
                    has_accum: bool) -> None:
    """Completeness marker (atomic): a round directory is a valid
    restore point only when the manifest exists AND every shard file it
    names is present.  Every rank writes the same content, so concurrent
    writers are idempotent (each stages through its own tmp name)."""
    import threading
    tmp = os.path.join(
        directory,
        f"{MANIFEST}.{os.getpid()}.{threading.get_ident()}.tmp")
    with open(tmp, "w") as f:
        json.dump({"n_shards": int(n_shards), "dim": int(dim),
                   "has_accum": bool(has_accum)}, f)
        f.flush()
        os.fsync(f.fileno())
    os.replace(tmp, os.path.join(directory, MANIFEST))


def _complete(directory: str) -> bool:
    """True iff the manifest exists and every shard file it promises is
    present (all writes are atomic, so present == whole).  No manifest
    == not a restore point — a crash before any rank finished its first
    shard must not be mistaken for a dump."""
    mpath = os.path.join(directory, MANIFEST)
    try:
        with open(mpath) as f:
            m = json.load(f)
    except (FileNotFoundError, ValueError, OSError):
        return False
    for s in range(int(m["n_shards"])):
        p = shard_path(directory, s)
        if not os.path.exists(p):
            return False
        if m.get("has_accum") and not os.path.exists(p + ".accum"):
            return False
    return True


def _read_one(path: str) -> Tuple[torch.Tensor, torch.Tensor]:
    import numpy as np
    with open(path, "rb") as f:
        if f.read(8) != MAGIC:
            raise ValueError(f"{path}: bad checkpoint magic")
        dim, dtype, count = struct.unpack("<qqq", f.read(24))
        if dtype != 0:
            raise ValueError(f"{path}: unsupported dtype {dtype}")
        ids = np.frombuffer(f.read(count * 8), dtype=np.int64)
        vals = np.frombuffer(f.read(count * dim * 4), dtype=np.float32)
    return (torch.from_numpy(ids.copy()),
            torch.from_numpy(vals.copy()).reshape(count, dim))


def shard_path(directory: str, shard: int, accum: bool = False) -> str:
    suffix = ".accum" if accum else ""
    return os.path.join(directory, f"shard_{shard:05d}.fps{suffix}")


def save_shard(directory: str, table: ShardedTable) -> None:
    """Write this shard's dump atomically, then (re)write the directory
    manifest.  The manifest names the full shard set, so a reader (or
    :func:`latest_checkpoint`) treats the directory as a restore point
    only once EVERY shard file exists — a crash mid-dump, or between
    ranks dumping into the shared round directory, leaves the previous
    complete dump as the newest valid one."""
    os.makedirs(directory, exist_ok=True)
    ids, values = table.dump()
    _write_one(shard_path(directory, table.shard), ids, values)
    if table.accum is not None:
        _write_one(shard_path(directory, table.shard, accum=True),
                   ids, table.accum)
    _write_manifest(directory, table.part.n_shards, table.dim,
                    table.accum is not None)


class PeriodicCheckpointer:
    """Coarse-recovery checkpoints: every
    ``every_rounds`` engine rounds each shard dumps itself under
    ``directory/round_NNNNNNNN/``; a restarted job resumes from
    :func:`latest_checkpoint` via :func:`load_shard`.  (The reference has
    NO recovery — Flink checkpoints don't cover iterative streams, its
    jobs are fail-fast; this is a new capability built on the reference's
    dump/load layout contract.)

    Pass to ``PSEngine(checkpointer=...)``; the engine calls
    ``maybe(round_idx)`` once per round after the server apply phase, so
    a dump always captures a round boundary (shard state contains every
    push the servers have applied; in-flight answers are re-pulled by the
    restarted workers — same recovery semantics as a model-load warm
    start).
    """

    def __init__(self, directory: str, table: ShardedTable,
                 every_rounds: int):
        if every_rounds < 1:
            raise ValueError("every_rounds must be >= 1")
        self.directory = directory
        self.table = table
        self.every_rounds = int(every_rounds)
        self.saved: List[str] = []

    def maybe(self, round_idx: int) -> None:
        if round_idx % self.every_rounds:
            return
        d = os.path.join(self.directory, f"round_{round_idx:08d}")
        save_shard(d, self.table)
        self.saved.append(d)


def latest_checkpoint(directory: str) -> Optional[str]:
    """Newest COMPLETE round_* subdirectory (manifest present and every
    shard file it names exists); incomplete/torn dumps are skipped."""
    if not os.path.isdir(directory):
        return None
    rounds = sorted(d for d in os.listdir(directory)
                    if d.startswith("round_"))
    for d in reversed(rounds):
        full = os.path.join(directory, d)
        if _complete(full):
            return full
    return None


def load_shard(directory: str, table: ShardedTable) -> None:
    """Load this shard's rows from a checkpoint written with ANY shard
    count: reads every file, keeps the ids this shard owns."""
    files = sorted(f for f in os.listdir(directory) if f.endswith(".fps"))
    if not files:
        raise FileNotFoundError(f"no .fps files in {directory}")
    for fname in files:
        ids, values = _read_one(os.path.join(directory, fname))
        mine = table.part.shard_of(ids) == table.shard
        if bool(mine.any()):
            table.load(ids[mine], values[mine])
        if table.accum is not None:
            apath = os.path.join(directory, fname + ".accum")
            if os.path.exists(apath):
                aids, avals = _read_one(apath)
                amine = table.part.shard_of(aids) == table.shard
                if bool(amine.any()):
                    rows = table.part.local_row(aids[amine]).to(
                        table.device)
                    table.accum.index_copy_(
                        0, rows, avals[amine].to(table.device))

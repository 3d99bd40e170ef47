"""Neuroglancer precomputed sharded format (neuroglancer_uint64_sharded_v1).

From-scratch restatement of the PUBLISHED neuroglancer sharded-format
specification (the reference outsources this to the cloud-volume
`ShardingSpecification` / `synthesize_shard_files` and the
`shard-computer` package, neither of which has source offline — see
/root/reference/igneous/task_creation/mesh.py:647-664,706-813 and
igneous/tasks/mesh/multires.py:26,388,497-508 for the call sites this
replaces).

Spec summary (little-endian throughout):
  hashed = hash(chunk_id >> preshift_bits)
  minishard number = hashed & (2^minishard_bits - 1)
  shard number    = (hashed >> minishard_bits) & (2^shard_bits - 1)
  shard file      = "<shard number as lowercase hex, zero-padded to
                     ceil(shard_bits/4) digits>.shard"
  shard file layout:
    [shard index: 2^minishard_bits pairs of uint64le (start, end) byte
     offsets of each minishard's index, relative to the END of the
     shard index]
    [chunk data] [minishard indices]
  minishard index ('raw'; 'gzip' wraps these bytes): 3*n uint64le —
    n chunk-id deltas (first absolute), n start-offset deltas (first
    relative to the end of the shard index, then each relative to the
    END of the previous chunk), n chunk byte sizes.
  hash 'murmurhash3_x86_128': low 8 bytes of MurmurHash3_x86_128
    (seed 0) of the uint64le chunk id.

`data_offset` mirrors the cloud-volume extension the reference relies
on for multires meshes (multires.py:388): when given, the recorded byte
range for a label covers only the TRAILING data_offset[label] bytes of
its value (the multilod manifest), with the fragment data stored in the
bytes immediately preceding it, exactly as the neuroglancer multires
spec requires.
"""
from __future__ import annotations

import gzip
from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np


# ---------------------------------------------------------------------------
# MurmurHash3_x86_128 (Austin Appleby's public-domain algorithm),
# vectorized over arrays of uint64 keys; returns the low 8 bytes of the
# 128-bit digest as uint64 (h1 | h2 << 32).

def _fmix32(h):
    h = h.astype(np.uint32, copy=True)
    h ^= h >> np.uint32(16)
    h *= np.uint32(0x85EBCA6B)
    h ^= h >> np.uint32(13)
    h *= np.uint32(0xC2B2AE35)
    h ^= h >> np.uint32(16)
    return h


def _rotl32(x, r):
    return ((x << np.uint32(r)) | (x >> np.uint32(32 - r))).astype(np.uint32)


def murmurhash3_x86_128_low64(keys) -> np.ndarray:
    """Low 64 bits of MurmurHash3_x86_128(uint64le bytes of key, seed=0)."""
    keys = np.asarray(keys, dtype=np.uint64)
    scalar = keys.ndim == 0
    keys = np.atleast_1d(keys)
    c1 = np.uint32(0x239B961B)
    c2 = np.uint32(0xAB0E9789)
    c3 = np.uint32(0x38B34AE5)
    # 8-byte input: no 16-byte blocks, all 8 bytes are tail (cases 8..1)
    h1 = np.zeros(len(keys), dtype=np.uint32)
    h2 = np.zeros(len(keys), dtype=np.uint32)
    h3 = np.zeros(len(keys), dtype=np.uint32)
    h4 = np.zeros(len(keys), dtype=np.uint32)
    k1 = (keys & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    k2 = (keys >> np.uint64(32)).astype(np.uint32)
    # tail cases 8..5 -> k2
    k2 = (k2 * c2).astype(np.uint32)
    k2 = _rotl32(k2, 16)
    k2 = (k2 * c3).astype(np.uint32)
    h2 ^= k2
    # tail cases 4..1 -> k1
    k1 = (k1 * c1).astype(np.uint32)
    k1 = _rotl32(k1, 15)
    k1 = (k1 * c2).astype(np.uint32)
    h1 ^= k1
    # finalization (len = 8)
    n8 = np.uint32(8)
    h1 ^= n8; h2 ^= n8; h3 ^= n8; h4 ^= n8
    h1 += h2 + h3 + h4
    h2 += h1; h3 += h1; h4 += h1
    h1 = _fmix32(h1); h2 = _fmix32(h2); h3 = _fmix32(h3); h4 = _fmix32(h4)
    h1 += h2 + h3 + h4
    h2 += h1  # h3, h4 not needed for the low 64 bits
    out = h1.astype(np.uint64) | (h2.astype(np.uint64) << np.uint64(32))
    return out[0] if scalar else out


# ---------------------------------------------------------------------------

@dataclass
class ShardingSpecification:
    type: str = "neuroglancer_uint64_sharded_v1"
    preshift_bits: int = 0
    hash: str = "murmurhash3_x86_128"
    minishard_bits: int = 0
    shard_bits: int = 0
    minishard_index_encoding: str = "raw"
    data_encoding: str = "raw"

    def to_dict(self) -> dict:
        return {
            "@type": self.type,
            "preshift_bits": int(self.preshift_bits),
            "hash": self.hash,
            "minishard_bits": int(self.minishard_bits),
            "shard_bits": int(self.shard_bits),
            "minishard_index_encoding": self.minishard_index_encoding,
            "data_encoding": self.data_encoding,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "ShardingSpecification":
        return cls(
            type=d.get("@type", "neuroglancer_uint64_sharded_v1"),
            preshift_bits=int(d.get("preshift_bits", 0)),
            hash=d.get("hash", "identity"),
            minishard_bits=int(d.get("minishard_bits", 0)),
            shard_bits=int(d.get("shard_bits", 0)),
            minishard_index_encoding=d.get("minishard_index_encoding",
                                           "raw"),
            data_encoding=d.get("data_encoding", "raw"),
        )

    # -- key routing --------------------------------------------------
    def hashed(self, labels) -> np.ndarray:
        labels = np.asarray(labels, dtype=np.uint64)
        shifted = labels >> np.uint64(self.preshift_bits)
        if self.hash == "identity":
            return shifted
        if self.hash == "murmurhash3_x86_128":
            return murmurhash3_x86_128_low64(shifted)
        raise ValueError(f"unknown shard hash {self.hash!r}")

    @property
    def shard_number_width(self) -> int:
        return max(int(np.ceil(self.shard_bits / 4.0)), 1)

    def shard_and_minishard(self, labels):
        h = np.atleast_1d(self.hashed(labels))
        mini = h & np.uint64((1 << self.minishard_bits) - 1)
        shard = (h >> np.uint64(self.minishard_bits)) & \
            np.uint64((1 << self.shard_bits) - 1)
        return shard, mini

    def compute_shard_location(self, label) -> str:
        shard, _ = self.shard_and_minishard([int(label)])
        return format(int(shard[0]), "x").zfill(self.shard_number_width)


def compute_shard_params_for_hashed(
        num_labels: int,
        shard_index_bytes: int = 2 ** 13,
        minishard_index_bytes: int = 2 ** 15,
        min_shards: int = 1):
    """Mirror of the reference's parameter solver
    (igneous/task_creation/common.py:140-213): balances shard-index and
    minishard-index sizes for uniformly hashed keys.
    Returns (shard_bits, minishard_bits, preshift_bits)."""
    assert min_shards >= 1
    if num_labels <= 0:
        return (0, 0, 0)

    num_minishards_per_shard = shard_index_bytes / 2 / 8
    labels_per_minishard = minishard_index_bytes / 3 / 8
    labels_per_shard = num_minishards_per_shard * labels_per_minishard

    if num_labels >= labels_per_shard:
        minishard_bits = np.ceil(np.log2(num_minishards_per_shard))
        shard_bits = np.ceil(np.log2(
            num_labels / (labels_per_minishard * (2 ** minishard_bits))))
    elif num_labels >= labels_per_minishard:
        minishard_bits = np.ceil(np.log2(num_labels / labels_per_minishard))
        shard_bits = 0
    else:
        minishard_bits = 0
        shard_bits = 0

    capacity = labels_per_shard * (2 ** shard_bits)
    utilized_capacity = num_labels / capacity
    if utilized_capacity <= 0.55:
        shard_bits -= 1

    shard_bits = max(shard_bits, 0)
    min_shard_bits = np.round(np.log2(min_shards))
    delta = max(min_shard_bits - shard_bits, 0)
    shard_bits += delta
    minishard_bits -= delta
    shard_bits = max(shard_bits, min_shard_bits)
    minishard_bits = max(minishard_bits, 0)
    return (int(shard_bits), int(minishard_bits), 0)


def assign_labels_to_shards(labels, preshift_bits: int, shard_bits: int,
                            minishard_bits: int,
                            hash: str = "murmurhash3_x86_128"
                            ) -> Dict[str, list]:
    """shard-computer equivalent: {shard_name: [labels...]} (reference
    call sites task_creation/mesh.py:664,764; multires.py:501-508)."""
    spec = ShardingSpecification(
        preshift_bits=preshift_bits, shard_bits=shard_bits,
        minishard_bits=minishard_bits, hash=hash)
    labels = np.asarray(labels, dtype=np.uint64)
    shard, _ = spec.shard_and_minishard(labels)
    out: Dict[str, list] = {}
    width = spec.shard_number_width
    for s in np.unique(shard):
        name = format(int(s), "x").zfill(width)
        out[name] = sorted(int(x) for x in labels[shard == s])
    return out


# ---------------------------------------------------------------------------

def synthesize_shard_files(spec: ShardingSpecification,
                           data: Dict[int, bytes],
                           data_offset: Optional[Dict[int, int]] = None,
                           ) -> Dict[str, bytes]:
    """Build complete ".shard" files from {label: value bytes}."""
    data_offset = data_offset or {}
    labels = np.asarray(sorted(int(k) for k in data), dtype=np.uint64)
    if len(labels) == 0:
        return {}
    shard, mini = spec.shard_and_minishard(labels)
    width = spec.shard_number_width
    files = {}
    for s in np.unique(shard):
        name = format(int(s), "x").zfill(width) + ".shard"
        sel = shard == s
        files[name] = _synthesize_one_shard(
            spec, labels[sel], mini[sel], data, data_offset)
    return files


def _synthesize_one_shard(spec, labels, mini, data, data_offset) -> bytes:
    n_mini = 1 << spec.minishard_bits
    chunks = []          # payload bytes, in (minishard, label) order
    index_blobs = [b""] * n_mini
    # (start, end) per minishard, filled after sizes are known
    shard_index = np.zeros((n_mini, 2), dtype="<u8")

    pos = 0  # offset relative to the end of the shard index
    per_mini = []
    for m in range(n_mini):
        msel = mini == np.uint64(m)
        mlabels = labels[msel]
        if len(mlabels) == 0:
            per_mini.append(None)
            continue
        ids = np.zeros(len(mlabels), dtype="<u8")
        starts = np.zeros(len(mlabels), dtype="<u8")
        sizes = np.zeros(len(mlabels), dtype="<u8")
        prev_id = 0
        prev_end = 0  # first offset is absolute (rel. to shard-index end)
        for i, label in enumerate(int(x) for x in mlabels):
            value = data[label]
            if spec.data_encoding == "gzip":
                value = gzip.compress(value, mtime=0)
            elif spec.data_encoding != "raw":
                raise ValueError(
                    f"unknown data_encoding {spec.data_encoding!r}")
            doff = data_offset.get(label)
            if doff is None:
                rec_start, rec_size = pos, len(value)
            else:
                # byte range covers only the trailing doff bytes (the
                # multilod manifest); fragments precede it
                rec_start = pos + len(value) - int(doff)
                rec_size = int(doff)
            ids[i] = label - prev_id
            starts[i] = rec_start - prev_end
            sizes[i] = rec_size
            prev_id = label
            prev_end = rec_start + rec_size
            chunks.append(value)
            pos += len(value)
        per_mini.append((ids, starts, sizes))

    data_len = pos
    for m in range(n_mini):
        if per_mini[m] is None:
            continue
        ids, starts, sizes = per_mini[m]
        blob = ids.tobytes() + starts.tobytes() + sizes.tobytes()
        if spec.minishard_index_encoding == "gzip":
            blob = gzip.compress(blob, mtime=0)
        index_blobs[m] = blob

    off = data_len
    for m in range(n_mini):
        blob = index_blobs[m]
        if blob:
            shard_index[m] = (off, off + len(blob))
            off += len(blob)
    return shard_index.tobytes() + b"".join(chunks) + b"".join(index_blobs)


# ---------------------------------------------------------------------------

class ShardReader:
    """Read side, for tests and the merge pipeline's consumers."""

    def __init__(self, spec: ShardingSpecification,
                 fetch):  # fetch(shard_filename) -> bytes
        self.spec = spec
        self.fetch = fetch

    def get(self, label: int) -> Optional[bytes]:
        """Return the recorded byte range for `label` (the manifest for
        multires meshes), or None if absent."""
        rng = self.byte_range(label)
        if rng is None:
            return None
        blob, start, size = rng
        return blob[start:start + size]

    def byte_range(self, label: int):
        spec = self.spec
        shard, mini = spec.shard_and_minishard([int(label)])
        name = format(int(shard[0]), "x").zfill(spec.shard_number_width) \
            + ".shard"
        blob = self.fetch(name)
        if blob is None:
            return None
        n_mini = 1 << spec.minishard_bits
        index = np.frombuffer(blob, dtype="<u8", count=2 * n_mini)
        idx_end = 16 * n_mini
        lo, hi = int(index[2 * int(mini[0])]), int(index[2 * int(mini[0]) + 1])
        if lo == hi:
            return None
        mindex = blob[idx_end + lo:idx_end + hi]
        if spec.minishard_index_encoding == "gzip":
            mindex = gzip.decompress(mindex)
        arr = np.frombuffer(mindex, dtype="<u8")
        n = len(arr) // 3
        ids = np.cumsum(arr[:n].astype(np.uint64))
        starts_delta = arr[n:2 * n].astype(np.int64)
        sizes = arr[2 * n:].astype(np.int64)
        # reconstruct absolute starts: each delta is relative to the end
        # of the previous chunk
        starts = np.zeros(n, dtype=np.int64)
        run = 0
        for i in range(n):
            starts[i] = run + starts_delta[i]
            run = starts[i] + sizes[i]
        hit = np.nonzero(ids == np.uint64(int(label)))[0]
        if len(hit) == 0:
            return None
        i = int(hit[0])
        if self.spec.data_encoding == "gzip":
            raw = blob[idx_end + starts[i]: idx_end + starts[i] + sizes[i]]
            out = gzip.decompress(raw)
            return (out, 0, len(out))
        return (blob, idx_end + int(starts[i]), int(sizes[i]))

    def list_labels_in_shard(self, shard_name: str):
        blob = self.fetch(shard_name + ".shard"
                          if not shard_name.endswith(".shard")
                          else shard_name)
        if blob is None:
            return []
        spec = self.spec
        n_mini = 1 << spec.minishard_bits
        index = np.frombuffer(blob, dtype="<u8", count=2 * n_mini)
        idx_end = 16 * n_mini
        labels = []
        for m in range(n_mini):
            lo, hi = int(index[2 * m]), int(index[2 * m + 1])
            if lo == hi:
                continue
            mindex = blob[idx_end + lo:idx_end + hi]
            if spec.minishard_index_encoding == "gzip":
                mindex = gzip.decompress(mindex)
            arr = np.frombuffer(mindex, dtype="<u8")
            n = len(arr) // 3
            labels.extend(int(x) for x in
                          np.cumsum(arr[:n].astype(np.uint64)))
        return sorted(labels)

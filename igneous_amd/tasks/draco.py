"""Draco quantization settings — mirror of
/root/reference/igneous/tasks/mesh/draco.py:7-59: computes the
quantization bits/range/origin for a chunk from its physical extent so
that the draco bin size stays below the voxel pitch."""
from __future__ import annotations

from typing import Any, Dict

import numpy as np


def draco_encoding_settings(shape, offset, resolution,
                            compression_level: int,
                            create_metadata: bool,
                            uses_new_draco_bin_size: bool = False
                            ) -> Dict[str, Any]:
    shape = np.asarray(shape)
    offset = np.asarray(offset)
    resolution = np.asarray(resolution)
    chunk_offset_nm = offset * resolution

    min_quantization_range = max(shape * resolution)
    if uses_new_draco_bin_size:
        max_draco_bin_size = np.floor(min(resolution) / 2)
    else:
        max_draco_bin_size = np.floor(min(resolution) / np.sqrt(2))

    (bits, qrange, bin_size) = calculate_draco_quantization_bits_and_range(
        min_quantization_range, max_draco_bin_size)
    quantization_origin = chunk_offset_nm - (chunk_offset_nm % bin_size)
    return {
        "quantization_bits": bits,
        "compression_level": compression_level,
        "quantization_range": qrange,
        "quantization_origin": quantization_origin,
        "create_metadata": create_metadata,
    }


def calculate_draco_quantization_bits_and_range(
        min_quantization_range, max_draco_bin_size,
        draco_quantization_bits=None) -> tuple:
    if draco_quantization_bits is None:
        draco_quantization_bits = np.ceil(
            np.log2(min_quantization_range / max_draco_bin_size + 1))
    num_draco_bins = 2 ** draco_quantization_bits - 1
    draco_bin_size = np.ceil(min_quantization_range / num_draco_bins)
    draco_quantization_range = draco_bin_size * num_draco_bins
    if draco_quantization_range < min_quantization_range + draco_bin_size:
        if draco_bin_size == max_draco_bin_size:
            return calculate_draco_quantization_bits_and_range(
                min_quantization_range, max_draco_bin_size,
                draco_quantization_bits + 1)
        else:
            draco_bin_size = draco_bin_size + 1
            draco_quantization_range = (draco_quantization_range
                                        + num_draco_bins)
    return draco_quantization_bits, draco_quantization_range, draco_bin_size

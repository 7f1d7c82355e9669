"""Incremental brotli decompression over the system libbrotlidec via
ctypes (no Python brotli package in this image; the shared library ships
with the OS). Mirrors the zlib.decompressobj interface the upstream
client uses so `content-encoding: br` responses re-chunk exactly like
gzip/deflate (reference: internal/extproc/util.go:15,57 decodes br with
andybalholm/brotli)."""

from __future__ import annotations

import ctypes
import ctypes.util
from typing import Optional

_BROTLI_RESULT_ERROR = 0
_BROTLI_RESULT_SUCCESS = 1
_BROTLI_RESULT_NEEDS_MORE_INPUT = 2
_BROTLI_RESULT_NEEDS_MORE_OUTPUT = 3

_lib: Optional[ctypes.CDLL] = None
_load_err: Optional[str] = None


def _load() -> ctypes.CDLL:
    global _lib, _load_err
    if _lib is not None:
        return _lib
    if _load_err is not None:
        raise BrotliUnavailable(_load_err)
    name = ctypes.util.find_library("brotlidec") or "libbrotlidec.so.1"
    try:
        lib = ctypes.CDLL(name)
        lib.BrotliDecoderCreateInstance.restype = ctypes.c_void_p
        lib.BrotliDecoderCreateInstance.argtypes = [ctypes.c_void_p] * 3
        lib.BrotliDecoderDestroyInstance.argtypes = [ctypes.c_void_p]
        lib.BrotliDecoderDecompressStream.restype = ctypes.c_int
        lib.BrotliDecoderDecompressStream.argtypes = [
            ctypes.c_void_p,
            ctypes.POINTER(ctypes.c_size_t),
            ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)),
            ctypes.POINTER(ctypes.c_size_t),
            ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)),
            ctypes.POINTER(ctypes.c_size_t),
        ]
        lib.BrotliDecoderIsFinished.restype = ctypes.c_int
        lib.BrotliDecoderIsFinished.argtypes = [ctypes.c_void_p]
    except OSError as e:  # pragma: no cover - library ships with the image
        _load_err = f"libbrotlidec unavailable: {e}"
        raise BrotliUnavailable(_load_err) from e
    _lib = lib
    return lib


class BrotliError(ValueError):
    pass


class BrotliUnavailable(RuntimeError):
    pass


def available() -> bool:
    try:
        _load()
        return True
    except BrotliUnavailable:
        return False


class BrotliDecompressor:
    """zlib.decompressobj-shaped incremental brotli decoder."""

    _CHUNK = 262144

    def __init__(self):
        lib = _load()
        self._lib = lib
        self._state = lib.BrotliDecoderCreateInstance(None, None, None)
        if not self._state:
            raise BrotliError("BrotliDecoderCreateInstance failed")
        self._finished = False

    def decompress(self, data: bytes) -> bytes:
        if self._state is None:
            raise BrotliError("decompressor closed")
        if self._finished or not data:
            return b""
        lib = self._lib
        out_parts = []
        in_buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
        avail_in = ctypes.c_size_t(len(data))
        next_in = ctypes.cast(in_buf, ctypes.POINTER(ctypes.c_uint8))
        while True:
            out_buf = (ctypes.c_uint8 * self._CHUNK)()
            avail_out = ctypes.c_size_t(self._CHUNK)
            next_out = ctypes.cast(out_buf, ctypes.POINTER(ctypes.c_uint8))
            rc = lib.BrotliDecoderDecompressStream(
                self._state, ctypes.byref(avail_in), ctypes.byref(next_in),
                ctypes.byref(avail_out), ctypes.byref(next_out), None,
            )
            produced = self._CHUNK - avail_out.value
            if produced:
                out_parts.append(bytes(out_buf[:produced]))
            if rc == _BROTLI_RESULT_ERROR:
                raise BrotliError("malformed brotli stream")
            if rc == _BROTLI_RESULT_NEEDS_MORE_OUTPUT:
                continue
            if rc == _BROTLI_RESULT_SUCCESS:
                self._finished = True
            break
        return b"".join(out_parts)

    def flush(self) -> bytes:
        # brotli produces output eagerly during decompress; nothing buffers
        return b""

    @property
    def eof(self) -> bool:
        return self._finished

    def __del__(self):
        if getattr(self, "_state", None):
            self._lib.BrotliDecoderDestroyInstance(self._state)
            self._state = None

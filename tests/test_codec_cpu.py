"""Pin the from-scratch raw-snappy decoder (codec.cpp) against pyarrow's
snappy COMPRESSOR (an independent implementation) on CPU."""

import ctypes

import numpy as np
import pyarrow as pa
import pytest

from paimon_amd import LIB_PATH, load_lib


def _snappy_dec(comp: bytes, cap: int) -> bytes:
    load_lib()
    lib = ctypes.CDLL(LIB_PATH)
    lib.pmh_debug_snappy.restype = ctypes.c_int64
    lib.pmh_debug_snappy.argtypes = [ctypes.c_char_p, ctypes.c_int64,
                                     ctypes.c_void_p, ctypes.c_int64]
    out = ctypes.create_string_buffer(cap)
    n = lib.pmh_debug_snappy(comp, len(comp), out, cap)
    if n < 0:
        lib.pmh_last_error.restype = ctypes.c_char_p
        raise RuntimeError(lib.pmh_last_error().decode())
    return out.raw[:n]


class TestSnappyDecoder:
    def test_roundtrip_payloads(self):
        rng = np.random.default_rng(11)
        payloads = [
            b"",
            b"a",
            b"hello hello hello hello" * 200,       # copy-heavy
            rng.integers(0, 256, 100_000, dtype=np.uint8).tobytes(),  # raw
            (b"abcd" * 17)[:61],                     # 61-byte literal edge
            bytes(rng.integers(0, 4, 300_000, dtype=np.uint8)),  # long copies
        ]
        for i, p in enumerate(payloads):
            comp = pa.compress(p, codec="snappy", asbytes=True)
            got = _snappy_dec(comp, len(p) + 16)
            assert got == p, f"payload {i}"

    def test_corrupt_input_rejected(self):
        with pytest.raises(RuntimeError):
            _snappy_dec(b"\xff\xff\xff\xff\xff\x00\x01\x02", 64)

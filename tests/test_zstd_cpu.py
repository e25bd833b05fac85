"""The from-scratch zstd frame decoder (paimon_amd/csrc/zstd_core.h, RFC 8878
restatement) against libzstd-produced frames via pyarrow's zstd codec.

This is the parity pin for the GPU page decoder k_zstd_pages, which runs the
SAME scalar core per page (tests/test_zstd_gpu.py covers the device path;
the reference delegates page decompression to parquet-java/aircompressor —
behaviourally, bytes out must equal ZSTD_decompress bytes out)."""
import numpy as np
import pyarrow as pa
import pytest

from paimon_amd.reader import debug_zstd_cpu


def _roundtrip(raw, level=3):
    comp = pa.Codec("zstd", compression_level=level).compress(raw)
    got = debug_zstd_cpu(comp.to_pybytes(), cap=len(raw) + 16)
    assert got == bytes(raw), (len(got), len(raw))


class TestScalarZstd:
    def test_empty(self):
        _roundtrip(b"")

    def test_tiny(self):
        _roundtrip(b"x")
        _roundtrip(b"hello zstd")

    def test_rle_like(self):
        _roundtrip(bytes(300_000))  # multi-block, RLE-heavy

    def test_text_repeats(self):
        _roundtrip(b"hello world " * 20_000)  # matches + repeat offsets

    def test_incompressible(self):
        rng = np.random.default_rng(1)
        _roundtrip(rng.integers(0, 256, 200_000, dtype=np.uint8).tobytes())

    def test_structured_ints(self):
        _roundtrip(np.arange(200_000, dtype=np.int32).tobytes())
        _roundtrip((np.arange(50_000, dtype=np.int64) * 7 + 3).tobytes())

    def test_levels(self):
        raw = np.sort(np.random.default_rng(2).integers(
            0, 1 << 20, 100_000, dtype=np.int64)).tobytes()
        for lvl in (1, 2, 3, 5, 9, 13, 19):
            _roundtrip(raw, level=lvl)

    def test_multiblock_4mb(self):
        rng = np.random.default_rng(3)
        raw = np.cumsum(rng.integers(0, 9, 600_000, dtype=np.int64)).tobytes()
        assert len(raw) > 4 << 20  # > 36 blocks of 128 KB
        _roundtrip(raw)

    def test_fuzz(self):
        rng = np.random.default_rng(4)
        for t in range(40):
            kind = t % 5
            size = int(rng.integers(1, 500_000))
            if kind == 0:
                raw = rng.integers(0, 3, size, dtype=np.uint8).tobytes()
            elif kind == 1:
                raw = rng.integers(0, 256, size, dtype=np.uint8).tobytes()
            elif kind == 2:
                raw = (b"abcdef" * (size // 6 + 1))[:size]
            elif kind == 3:
                raw = np.sort(rng.integers(0, 1 << 20, size // 4 + 1,
                                           dtype=np.int32)).tobytes()[:size]
            else:
                raw = bytes(np.where(rng.random(size) < 0.9, 42,
                                     rng.integers(0, 256, size)
                                     ).astype(np.uint8))
            _roundtrip(raw, level=int(rng.choice([1, 3, 9])))

    def test_corrupt_rejected(self):
        comp = bytearray(pa.Codec("zstd").compress(
            b"some compressible payload " * 100).to_pybytes())
        comp[0] ^= 0xFF  # break the magic
        with pytest.raises(RuntimeError):
            debug_zstd_cpu(bytes(comp), cap=1 << 16)


class TestNoContentSizeFrames:
    """Streaming-written frames omit the frame content size (FCS) —
    pz_decode_frame must decode them (parquet pages always carry FCS, but
    the decoder follows RFC 8878, not the writer's habits)."""

    def test_streaming_frame(self):
        import io
        raw = (b"streaming frame payload " * 4000)
        buf = pa.BufferOutputStream()
        with pa.CompressedOutputStream(buf, "zstd") as z:
            z.write(raw)
        comp = buf.getvalue().to_pybytes()
        got = debug_zstd_cpu(comp, cap=len(raw) + 64)
        assert got == raw

    def test_streaming_incompressible(self):
        rng = np.random.default_rng(9)
        raw = rng.integers(0, 256, 300_000, dtype=np.uint8).tobytes()
        buf = pa.BufferOutputStream()
        with pa.CompressedOutputStream(buf, "zstd") as z:
            z.write(raw)
        comp = buf.getvalue().to_pybytes()
        got = debug_zstd_cpu(comp, cap=len(raw) + 64)
        assert got == raw

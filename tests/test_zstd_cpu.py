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


class TestEncoder:
    """The from-scratch zstd ENCODER (pz_encode_frame: greedy LZ + RAW
    literals + predefined-FSE sequences). Frames must decode with LIBZSTD
    (pyarrow) bit-exactly — that is the validity contract; ratio is
    workload-dependent and intentionally below libzstd's (no huffman
    literals, no optimal parse)."""

    def _rt(self, raw, min_ratio=None):
        from paimon_amd.reader import debug_zstd_enc_cpu
        comp = debug_zstd_enc_cpu(raw)
        back = pa.Codec("zstd").decompress(
            comp, decompressed_size=len(raw)).to_pybytes()
        assert back == bytes(raw)
        ours = debug_zstd_cpu(comp, cap=len(raw) + 16)  # our decoder too
        assert ours == bytes(raw)
        if min_ratio:
            assert len(raw) / len(comp) >= min_ratio, \
                (len(raw), len(comp))

    def test_empty_and_tiny(self):
        self._rt(b"")
        self._rt(b"x")
        self._rt(b"abcd" * 2)

    def test_text(self):
        self._rt(b"the quick brown fox jumps " * 20_000, min_ratio=50)

    def test_zeros_multiblock(self):
        self._rt(bytes(1 << 20), min_ratio=1000)

    def test_incompressible_random(self):
        rng = np.random.default_rng(21)
        self._rt(rng.integers(0, 256, 500_000, dtype=np.uint8).tobytes())

    def test_ascending_int64(self):
        # parquet key-column shape: high bytes repeat at stride 8
        self._rt(np.arange(500_000, dtype=np.int64).tobytes(),
                 min_ratio=2.0)

    def test_fuzz(self):
        from paimon_amd.reader import debug_zstd_enc_cpu
        rng = np.random.default_rng(22)
        codec = pa.Codec("zstd")
        for t in range(30):
            kind = t % 5
            size = int(rng.integers(0, 800_000))
            if kind == 0:
                raw = rng.integers(0, 3, size, dtype=np.uint8).tobytes()
            elif kind == 1:
                raw = rng.integers(0, 256, size, dtype=np.uint8).tobytes()
            elif kind == 2:
                raw = (b"pattern-xyz" * (size // 11 + 1))[:size]
            elif kind == 3:
                raw = np.cumsum(rng.integers(0, 7, size // 8 + 1,
                                             dtype=np.int64)
                                ).tobytes()[:size]
            else:
                raw = bytes(np.where(rng.random(size) < 0.85, 7,
                                     rng.integers(0, 256, size)
                                     ).astype(np.uint8))
            comp = debug_zstd_enc_cpu(raw)
            back = codec.decompress(
                comp, decompressed_size=len(raw)).to_pybytes()
            assert back == raw, (t, kind, size)


class TestCorruptionRobustness:
    """Bit-flipped / truncated frames must fail CLEANLY (error code) or —
    when the flip lands in an unchecked value region — produce output no
    larger than requested; never crash, hang, or overrun. (The staging
    batch verifies decompressed size == page header, so a wrong-but-clean
    decode falls back to the host codec.)"""

    def test_bit_flips(self):
        rng = np.random.default_rng(61)
        raw = np.cumsum(rng.integers(0, 7, 40_000, dtype=np.int64)
                        ).tobytes()
        comp = bytearray(pa.Codec("zstd", compression_level=5
                                  ).compress(raw).to_pybytes())
        for t in range(200):
            c = bytearray(comp)
            pos = int(rng.integers(0, len(c)))
            c[pos] ^= 1 << int(rng.integers(0, 8))
            try:
                out = debug_zstd_cpu(bytes(c), cap=len(raw) + 16)
                assert len(out) <= len(raw) + 16
            except RuntimeError:
                pass  # clean error is the expected outcome

    def test_truncations(self):
        raw = b"truncate me please " * 5000
        comp = pa.Codec("zstd").compress(raw).to_pybytes()
        for cut in range(0, len(comp), max(1, len(comp) // 64)):
            try:
                out = debug_zstd_cpu(comp[:cut], cap=len(raw) + 16)
                assert len(out) <= len(raw) + 16
            except RuntimeError:
                pass

    def test_random_garbage(self):
        rng = np.random.default_rng(62)
        for t in range(60):
            blob = rng.integers(0, 256, int(rng.integers(1, 5000)),
                                dtype=np.uint8).tobytes()
            try:
                debug_zstd_cpu(blob, cap=1 << 16)
            except RuntimeError:
                pass

    def test_encoder_deterministic(self):
        from paimon_amd.reader import debug_zstd_enc_cpu
        raw = np.arange(100_000, dtype=np.int64).tobytes()
        assert debug_zstd_enc_cpu(raw) == debug_zstd_enc_cpu(raw)

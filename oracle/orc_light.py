"""Minimal ORC file-structure parser (protobuf wire format) — TEST
INFRASTRUCTURE for locating column streams inside files written by
pyarrow.orc, so the oracle ORC decoders (and later the GPU RLEv2 kernels)
can be pinned stream-by-stream against pyarrow's independent reads.

The ORC RLEv2 arithmetic lives in the orc-core 1.9.8 dependency of the
reference (not vendored): parity is pinned at this boundary
via pyarrow.orc round trips + synthetic known-answer vectors.
"""

from dataclasses import dataclass, field
from typing import List

# protobuf wire types
_VARINT, _I64, _LEN, _I32 = 0, 1, 2, 5

COMP_NONE, COMP_ZLIB, COMP_SNAPPY, COMP_LZO, COMP_LZ4, COMP_ZSTD = range(6)

TK_BOOLEAN, TK_BYTE, TK_SHORT, TK_INT, TK_LONG, TK_FLOAT, TK_DOUBLE = range(7)
TK_STRING, TK_BINARY, TK_TIMESTAMP, TK_LIST, TK_MAP, TK_STRUCT = 7, 8, 9, 10, 11, 12

STREAM_PRESENT, STREAM_DATA, STREAM_LENGTH, STREAM_DICTIONARY = 0, 1, 2, 3

ENC_DIRECT, ENC_DICTIONARY, ENC_DIRECT_V2, ENC_DICTIONARY_V2 = 0, 1, 2, 3


def _uvarint(b, p):
    v, shift = 0, 0
    while True:
        x = b[p]
        p += 1
        v |= (x & 0x7F) << shift
        if not (x & 0x80):
            return v, p
        shift += 7


def _fields(buf):
    """Parse a protobuf message into {field_num: [values]} (values: int for
    varint/fixed, bytes for length-delimited)."""
    out = {}
    p = 0
    n = len(buf)
    while p < n:
        hdr, p = _uvarint(buf, p)
        fnum, wt = hdr >> 3, hdr & 7
        if wt == _VARINT:
            v, p = _uvarint(buf, p)
        elif wt == _LEN:
            ln, p = _uvarint(buf, p)
            v = bytes(buf[p:p + ln])
            p += ln
        elif wt == _I64:
            v = int.from_bytes(buf[p:p + 8], "little")
            p += 8
        elif wt == _I32:
            v = int.from_bytes(buf[p:p + 4], "little")
            p += 4
        else:
            raise ValueError(f"bad protobuf wire type {wt}")
        out.setdefault(fnum, []).append(v)
    return out


@dataclass
class OrcStream:
    kind: int
    column: int
    length: int
    offset: int = 0  # absolute file offset (filled by parse)


@dataclass
class OrcStripe:
    offset: int
    index_length: int
    data_length: int
    footer_length: int
    num_rows: int
    streams: List[OrcStream] = field(default_factory=list)
    encodings: List[int] = field(default_factory=list)


@dataclass
class OrcFileInfo:
    num_rows: int
    compression: int
    column_names: List[str]    # flat struct fields; col id = index + 1
    column_kinds: List[int]
    stripes: List[OrcStripe] = field(default_factory=list)


def parse_orc(path) -> OrcFileInfo:
    with open(path, "rb") as f:
        data = f.read()
    assert data[:3] == b"ORC", "not an ORC file"
    ps_len = data[-1]
    ps = _fields(data[-1 - ps_len:-1])
    footer_len = ps[1][0]
    compression = ps.get(2, [0])[0]
    if compression != COMP_NONE:
        raise NotImplementedError(
            "oracle ORC parser supports uncompressed files (write with "
            "compression='uncompressed')")
    fstart = len(data) - 1 - ps_len - footer_len
    footer = _fields(data[fstart:fstart + footer_len])
    # types: field 4 (repeated). Root is STRUCT with subtypes + fieldNames.
    types = [
        _fields(t) for t in footer.get(4, [])
    ]
    def _ints(vals):
        # repeated uint32 may arrive packed (length-delimited varint blob)
        out = []
        for v in vals:
            if isinstance(v, int):
                out.append(v)
            else:
                p = 0
                while p < len(v):
                    x, p = _uvarint(v, p)
                    out.append(x)
        return out

    root = types[0]
    names = [x.decode() for x in root.get(3, [])]
    kinds = [types[i].get(1, [0])[0] for i in _ints(root.get(2, []))]
    fi = OrcFileInfo(num_rows=footer.get(6, [0])[0], compression=compression,
                     column_names=names, column_kinds=kinds)
    for s in footer.get(3, []):
        sf = _fields(s)
        st = OrcStripe(offset=sf[1][0], index_length=sf.get(2, [0])[0],
                       data_length=sf[3][0], footer_length=sf[4][0],
                       num_rows=sf[5][0])
        # stripe footer
        sf_off = st.offset + st.index_length + st.data_length
        spf = _fields(data[sf_off:sf_off + st.footer_length])
        pos = st.offset
        for sb in spf.get(1, []):
            sm = _fields(sb)
            stream = OrcStream(kind=sm.get(1, [0])[0],
                               column=sm.get(2, [0])[0],
                               length=sm.get(3, [0])[0], offset=pos)
            pos += stream.length
            st.streams.append(stream)
        st.encodings = [_fields(e).get(1, [0])[0] for e in spf.get(2, [])]
        fi.stripes.append(st)
    return fi


def find_stream(stripe: OrcStripe, column: int, kind: int):
    for s in stripe.streams:
        if s.column == column and s.kind == kind:
            return s
    return None

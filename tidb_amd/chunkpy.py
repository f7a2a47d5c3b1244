"""numpy-backed reference-layout chunks (pkg/util/chunk layout).

Fixed columns: raw little-endian arrays (8 B ints/times/floats, 40 B decimals);
null bitmap 1 bit/row LSB-first, 1 = NOT NULL; var-len: offsets + bytes.
"""
import ctypes

import numpy as np

from tests.gxlib import (GxChunk, GxCol, GX_TYPE_DECIMAL, GX_TYPE_F64,
                         GX_TYPE_STRING)


class PyColumn:
    """Owns the buffers behind one gx_col."""

    def __init__(self, typ, n_rows_cap, data_cap_bytes=None, frac=0):
        self.typ = typ
        self.frac = frac
        if typ == GX_TYPE_STRING:
            if data_cap_bytes is None:
                data_cap_bytes = n_rows_cap * 16
            self.offsets = np.zeros(n_rows_cap + 1, dtype=np.int64)
            self.data = np.zeros(data_cap_bytes, dtype=np.uint8)
        else:
            elem = 40 if typ == GX_TYPE_DECIMAL else 8
            self.offsets = None
            self.data = np.zeros(n_rows_cap * elem, dtype=np.uint8)
        self.null_bitmap = np.full((n_rows_cap + 7) // 8, 0xFF, dtype=np.uint8)
        self.length = 0

    def as_gx(self):
        c = GxCol()
        c.data = self.data.ctypes.data_as(ctypes.c_void_p)
        c.null_bitmap = self.null_bitmap.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8))
        if self.offsets is not None:
            c.offsets = self.offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))
            c.offsets_cap = len(self.offsets)
            c.elem_size = -1
        else:
            c.offsets = None
            c.offsets_cap = 0
            c.elem_size = 40 if self.typ == GX_TYPE_DECIMAL else 8
        c.length = self.length
        c.data_cap = len(self.data)
        return c


class PyChunk:
    def __init__(self, col_types, n_rows_cap, col_fracs=None, data_caps=None):
        self.col_types = list(col_types)
        self.col_fracs = list(col_fracs) if col_fracs else [0] * len(col_types)
        self.columns = []
        for i, t in enumerate(col_types):
            cap = None if data_caps is None else data_caps[i]
            self.columns.append(PyColumn(t, n_rows_cap, cap, self.col_fracs[i]))
        self._gx_cols = (GxCol * len(self.columns))()
        self.chunk = GxChunk()

    def as_gx(self):
        for i, col in enumerate(self.columns):
            self._gx_cols[i] = col.as_gx()
        self.chunk.cols = self._gx_cols
        self.chunk.n_cols = len(self.columns)
        self.chunk.n_rows = self.columns[0].length if self.columns else 0
        return self.chunk

    # -- read back helpers (after gx_next fills the buffers) --
    def rows(self, n_rows):
        out = []
        for i in range(n_rows):
            out.append(tuple(self.cell(c, i) for c in range(len(self.columns))))
        return out

    def is_null(self, col, row):
        g = self._gx_cols[col]
        return (self.columns[col].null_bitmap[row // 8] >> (row % 8)) & 1 == 0

    def cell(self, col, row):
        from tidb_amd.decimals import decimal_bytes_to_str
        c = self.columns[col]
        if self.is_null(col, row):
            return None
        if c.typ == GX_TYPE_STRING:
            s, e = c.offsets[row], c.offsets[row + 1]
            return bytes(c.data[s:e]).decode("utf8", "replace")
        if c.typ == GX_TYPE_DECIMAL:
            return decimal_bytes_to_str(bytes(c.data[row * 40:(row + 1) * 40]))
        if c.typ == GX_TYPE_F64:
            import struct
            return struct.unpack("<d", bytes(c.data[row * 8:(row + 1) * 8]))[0]
        v = int.from_bytes(bytes(c.data[row * 8:(row + 1) * 8]), "little", signed=True)
        return v

    # -- fill helpers for input chunks --
    def append_row(self, values):
        """values: list matching col types — int (i64/time raw u64),
        bytes/str (string), bytes-of-40 (decimal), None (NULL)."""
        for c, v in zip(self.columns, values):
            i = c.length
            if v is None:
                c.null_bitmap[i // 8] &= ~(1 << (i % 8)) & 0xFF
                if c.offsets is not None:
                    c.offsets[i + 1] = c.offsets[i]
            else:
                c.null_bitmap[i // 8] |= 1 << (i % 8)
                if c.offsets is not None:
                    b = v.encode() if isinstance(v, str) else v
                    s = c.offsets[i]
                    c.data[s:s + len(b)] = np.frombuffer(b, dtype=np.uint8)
                    c.offsets[i + 1] = s + len(b)
                elif c.typ == GX_TYPE_DECIMAL:
                    c.data[i * 40:(i + 1) * 40] = np.frombuffer(v, dtype=np.uint8)
                elif c.typ == GX_TYPE_F64:
                    import struct
                    c.data[i * 8:(i + 1) * 8] = np.frombuffer(
                        struct.pack("<d", float(v)), dtype=np.uint8)
                else:
                    if v < 0:
                        v += 1 << 64
                    c.data[i * 8:(i + 1) * 8] = np.frombuffer(
                        int(v).to_bytes(8, "little"), dtype=np.uint8)
            c.length += 1

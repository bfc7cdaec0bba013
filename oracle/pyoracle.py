"""ctypes bindings for the CPU oracle (oracle/liboracle.so) and the
reference-pinned hash library (oracle/_ref/libpg_hashref.so).

TEST INFRASTRUCTURE ONLY: imported by tests/, __graft_entry__.smoke()'s
checker and bench.py's cpu_baseline leg — never by the product package.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))


class Q1Group(ctypes.Structure):
    _fields_ = [
        ("count", ctypes.c_int64),
        ("sum_qty_c", ctypes.c_int64),
        ("sum_base_c", ctypes.c_int64),
        ("sum_dcol_c", ctypes.c_int64),
        ("disc_lo", ctypes.c_uint64),
        ("disc_hi", ctypes.c_int64),
        ("charge_lo", ctypes.c_uint64),
        ("charge_hi", ctypes.c_int64),
    ]

    def as_dict(self):
        return {
            "count": self.count,
            "sum_qty_c": self.sum_qty_c,
            "sum_base_c": self.sum_base_c,
            "sum_dcol_c": self.sum_dcol_c,
            "sum_disc4": (self.disc_hi << 64) | self.disc_lo,
            "sum_charge6": (self.charge_hi << 64) | self.charge_lo,
        }


class Q3Row(ctypes.Structure):
    _fields_ = [
        ("orderkey", ctypes.c_int64),
        ("rev_lo", ctypes.c_uint64),
        ("rev_hi", ctypes.c_int64),
        ("orderdate", ctypes.c_int32),
        ("shippriority", ctypes.c_int32),
    ]

    def as_dict(self):
        return {
            "orderkey": self.orderkey,
            "revenue4": (self.rev_hi << 64) | self.rev_lo,
            "orderdate": self.orderdate,
            "shippriority": self.shippriority,
        }


class Q5Group(ctypes.Structure):
    _fields_ = [
        ("count", ctypes.c_int64),
        ("rev_lo", ctypes.c_uint64),
        ("rev_hi", ctypes.c_int64),
    ]

    def as_dict(self):
        return {"count": self.count,
                "revenue4": (self.rev_hi << 64) | self.rev_lo}


class Q3Result(ctypes.Structure):
    _fields_ = [
        ("n_out", ctypes.c_int64),
        ("n_groups", ctypes.c_int64),
        ("rev_sum_lo", ctypes.c_uint64),
        ("rev_sum_hi", ctypes.c_int64),
        ("group_checksum", ctypes.c_uint64),
        ("n_join_rows", ctypes.c_int64),
    ]

    def as_dict(self):
        return {
            "n_groups": self.n_groups,
            "rev_sum4": (self.rev_sum_hi << 64) | self.rev_sum_lo,
            "group_checksum": self.group_checksum,
            "n_join_rows": self.n_join_rows,
        }


def _build():
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


def _load(name):
    path = os.path.join(_DIR, name)
    if not os.path.exists(path):
        _build()
    return ctypes.CDLL(path)


_lib = None
_ref = None

I64 = ctypes.c_int64
I32 = ctypes.c_int32
U64 = ctypes.c_uint64
U32 = ctypes.c_uint32

_P_I64 = np.ctypeslib.ndpointer(np.int64, flags="C_CONTIGUOUS")
_P_I32 = np.ctypeslib.ndpointer(np.int32, flags="C_CONTIGUOUS")
_P_U8 = np.ctypeslib.ndpointer(np.uint8, flags="C_CONTIGUOUS")


def lib():
    global _lib
    if _lib is None:
        _lib = _load("liboracle.so")
        L = _lib
        L.gg_oracle_q1_arrays.restype = ctypes.c_int
        L.gg_oracle_q1_arrays.argtypes = [
            _P_I32, _P_U8, _P_U8, _P_I64, _P_I64, _P_I64, _P_I64,
            I64, I32, ctypes.POINTER(Q1Group), ctypes.c_int]
        L.gg_oracle_q1_synth.restype = ctypes.c_int
        L.gg_oracle_q1_synth.argtypes = [
            U64, I64, I64, I64, I32, ctypes.POINTER(Q1Group), ctypes.c_int]
        L.gg_oracle_q1_volcano_synth.restype = ctypes.c_int
        L.gg_oracle_q1_volcano_synth.argtypes = [
            U64, I64, I64, I64, I32, ctypes.POINTER(Q1Group), ctypes.c_int]
        L.gg_oracle_q1_synth_segment.restype = ctypes.c_int
        L.gg_oracle_q1_synth_segment.argtypes = [
            U64, I64, I32, I32, I32, ctypes.POINTER(Q1Group), ctypes.c_int]
        L.gg_numeric_to_str.argtypes = [U64, I64, ctypes.c_int, ctypes.c_char_p]
        L.gg_avg_rscale.restype = ctypes.c_int
        L.gg_avg_rscale.argtypes = [U64, I64, ctypes.c_int, I64]
        L.gg_numeric_avg_to_str.argtypes = [U64, I64, ctypes.c_int, I64,
                                            ctypes.c_char_p]
        L.gg_oracle_q3_arrays.restype = ctypes.c_int
        L.gg_oracle_q3_arrays.argtypes = [
            _P_I64, _P_U8, I64, ctypes.c_uint8,
            _P_I64, _P_I64, _P_I32, _P_I32, I64,
            _P_I64, _P_I32, _P_I64, _P_I64, I64,
            I32, I64, ctypes.POINTER(Q3Row), ctypes.POINTER(Q3Result),
            ctypes.c_int]
        L.gg_oracle_q3_synth.restype = ctypes.c_int
        L.gg_oracle_q3_synth.argtypes = [
            U64, I64, I32, I64, ctypes.POINTER(Q3Row),
            ctypes.POINTER(Q3Result), ctypes.c_int]
        L.gg_oracle_sumprice_arrays.restype = ctypes.c_int
        L.gg_oracle_sumprice_arrays.argtypes = [
            _P_I32, _P_I64, I64, I32, ctypes.POINTER(I64),
            ctypes.POINTER(I64), ctypes.c_int]
        L.gg_oracle_sumprice_synth.restype = ctypes.c_int
        L.gg_oracle_sumprice_synth.argtypes = [
            U64, I64, I32, ctypes.POINTER(I64), ctypes.POINTER(I64),
            ctypes.c_int]
        L.gg_oracle_hash_any.restype = U32
        L.gg_oracle_hash_any.argtypes = [ctypes.c_char_p, ctypes.c_int]
        L.gg_oracle_hash_uint32.restype = U32
        L.gg_oracle_hash_uint32.argtypes = [U32]
        L.gg_oracle_hashint4.restype = U32
        L.gg_oracle_hashint4.argtypes = [I32]
        L.gg_oracle_hashint8.restype = U32
        L.gg_oracle_hashint8.argtypes = [I64]
        L.gg_oracle_hashchar.restype = U32
        L.gg_oracle_hashchar.argtypes = [ctypes.c_char]
        L.gg_oracle_segment_int8.restype = I32
        L.gg_oracle_segment_int8.argtypes = [I64, I32]
        L.gg_oracle_segment_int4.restype = I32
        L.gg_oracle_segment_int4.argtypes = [I32, I32]
        L.gg_oracle_jump_hash.restype = I32
        L.gg_oracle_jump_hash.argtypes = [U64, I32]
        L.gg_oracle_pgdate.restype = I32
        L.gg_oracle_pgdate.argtypes = [ctypes.c_int] * 3
        L.gg_oracle_gen_lineitem.argtypes = [
            U64, I64, I64, _P_I64, _P_I64, _P_I64, _P_I64, _P_I64,
            _P_I32, _P_U8, _P_U8]
        L.gg_oracle_gen_orders.argtypes = [
            U64, I64, I64, I64, _P_I64, _P_I64, _P_I32, _P_I32]
        L.gg_oracle_gen_customer.argtypes = [U64, I64, I64, _P_I64, _P_U8,
                                             _P_U8]
        L.gg_oracle_gen_supplier.argtypes = [U64, I64, I64, _P_I64, _P_U8]
        L.gg_oracle_gen_l_suppkey.argtypes = [U64, I64, I64, I64, _P_I64]
        L.gg_oracle_q5_synth.restype = ctypes.c_int
        L.gg_oracle_q5_synth.argtypes = [U64, I64, I32, I32, I32,
                                         ctypes.POINTER(Q5Group),
                                         ctypes.c_int]
        L.gg_oracle_q5_arrays.restype = ctypes.c_int
        L.gg_oracle_q5_arrays.argtypes = [
            _P_I64, _P_U8, I64,
            _P_I64, _P_I64, _P_I32, I64,
            _P_I64, _P_I64, _P_I64, _P_I64, I64,
            _P_I64, _P_U8, I64,
            _P_I32, I32, I32, I32, ctypes.POINTER(Q5Group), ctypes.c_int]
    return _lib


def ref():
    """The reference's own hashfunc.c, compiled in place (None if the
    prebuilt .so is missing and /root/reference is absent)."""
    global _ref
    if _ref is None:
        path = os.path.join(_DIR, "_ref", "libpg_hashref.so")
        if not os.path.exists(path) and os.path.isdir("/root/reference"):
            _build()
        if not os.path.exists(path):
            return None
        _ref = ctypes.CDLL(path)
        R = _ref
        R.ref_hash_any.restype = U32
        R.ref_hash_any.argtypes = [ctypes.c_char_p, ctypes.c_int]
        R.ref_hash_uint32.restype = U32
        R.ref_hash_uint32.argtypes = [U32]
        R.ref_hashint4.restype = U32
        R.ref_hashint4.argtypes = [I32]
        R.ref_hashint8.restype = U32
        R.ref_hashint8.argtypes = [I64]
        R.ref_hashchar.restype = U32
        R.ref_hashchar.argtypes = [ctypes.c_char]
    return _ref


_dsb = None


_tupser = None


def tupser_ref():
    """The reference's own Motion chunk serializer (tupser.c
    SerializeTuple compiled in place); None when the prebuilt .so is
    missing and the reference tree is absent."""
    global _tupser
    if _tupser is None:
        path = os.path.join(_DIR, "_ref", "libpg_tupserref.so")
        if not os.path.exists(path) and os.path.isdir("/root/reference"):
            _build()
        if not os.path.exists(path):
            return None
        _tupser = ctypes.CDLL(path)
        _tupser.ref_tupser_chunks.restype = ctypes.c_int
        _tupser.ref_tupser_chunks.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_void_p, ctypes.c_int]
    return _tupser


def ref_tupser_chunks(memtuple_bytes, natts, max_chunk):
    """Chunk stream the REFERENCE SerializeTuple emits for one
    MemTuple (b->pri=NULL → the chunked interconnect path)."""
    T = tupser_ref()
    assert T is not None, "reference tupser codec missing"
    mt = np.ascontiguousarray(memtuple_bytes, np.uint8)
    cap = len(mt) + (len(mt) // 16 + 64) * 8 + (1 << 16)
    out = np.zeros(cap, np.uint8)
    n = T.ref_tupser_chunks(mt.ctypes.data_as(ctypes.c_void_p), len(mt),
                            natts, max_chunk,
                            out.ctypes.data_as(ctypes.c_void_p), cap)
    assert n >= 0, n
    return out[:n].copy()


def dsb_ref():
    """The reference's own AOCS datum-stream codec (datumstreamblock.c
    compiled in place); None when the prebuilt .so is missing and the
    reference tree is absent."""
    global _dsb
    if _dsb is None:
        path = os.path.join(_DIR, "_ref", "libpg_dsbref.so")
        if not os.path.exists(path) and os.path.isdir("/root/reference"):
            _build()
        if not os.path.exists(path):
            return None
        _dsb = ctypes.CDLL(path)
        D = _dsb
        D.ref_dsb_encode.restype = ctypes.c_int
        D.ref_dsb_encode.argtypes = [
            _P_I64, _P_U8, I64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, I32, _P_U8, I64, ctypes.POINTER(I64),
            ctypes.POINTER(ctypes.c_int32)]
        D.ref_dsb_decode.restype = ctypes.c_int
        D.ref_dsb_decode.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            _P_I64, _P_U8, I64, ctypes.POINTER(I64)]
        D.ref_ao_wrap_stream.restype = ctypes.c_int
        D.ref_ao_wrap_stream.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, _P_U8, I64,
            ctypes.POINTER(I64)]
        D.ref_dsb_encode_text.restype = ctypes.c_int
        D.ref_dsb_encode_text.argtypes = [
            _P_U8, _P_I64, _P_U8, I64, ctypes.c_int, ctypes.c_int,
            ctypes.c_int32, _P_U8, I64, ctypes.POINTER(I64),
            ctypes.POINTER(ctypes.c_int32)]
        D.ref_dsb_decode_text.restype = ctypes.c_int
        D.ref_dsb_decode_text.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, _P_U8, I64, _P_I64,
            _P_U8, I64, ctypes.POINTER(I64)]
        D.ref_ao_wrap_stream_c.restype = ctypes.c_int
        D.ref_ao_wrap_stream_c.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, _P_U8, I64, ctypes.POINTER(I64)]
        D.ref_mt_create_binding.restype = ctypes.c_void_p
        D.ref_mt_create_binding.argtypes = [
            ctypes.c_int, ctypes.POINTER(ctypes.c_int32), _P_U8,
            ctypes.c_char_p, ctypes.POINTER(ctypes.c_uint32)]
        D.ref_mt_form.restype = ctypes.c_int32
        D.ref_mt_form.argtypes = [
            ctypes.c_void_p, _P_I64, _P_U8, _P_U8, ctypes.c_int32]
        D.ref_mt_getattr.restype = ctypes.c_int32
        D.ref_mt_getattr.argtypes = [
            ctypes.c_void_p, _P_U8, ctypes.c_int, ctypes.POINTER(I64),
            ctypes.POINTER(ctypes.c_uint8)]
        D.ref_mt_form_var.restype = ctypes.c_int32
        D.ref_mt_form_var.argtypes = [
            ctypes.c_void_p, _P_I64, _P_U8, _P_U8, _P_I64, _P_U8, _P_U8,
            ctypes.c_int32]
        D.ref_mt_get_colbind.restype = ctypes.c_int32
        D.ref_mt_get_colbind.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int] + \
            [ctypes.POINTER(ctypes.c_int32)] * 6
        D.ref_mt_binding_info.restype = ctypes.c_int32
        D.ref_mt_binding_info.argtypes = [
            ctypes.c_void_p] + [ctypes.POINTER(ctypes.c_int32)] * 3
        D.ref_ao_wrap_stream_bd.restype = ctypes.c_int
        D.ref_ao_wrap_stream_bd.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int, _P_U8, I64, ctypes.POINTER(I64)]
        D.ref_ao_wrap_stream_large.restype = ctypes.c_int
        D.ref_ao_wrap_stream_large.argtypes = [
            _P_U8, I64, ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int32, _P_U8, I64, ctypes.POINTER(I64)]
        D.ref_ao_probe_block.restype = ctypes.c_int
        D.ref_ao_probe_block.argtypes = [
            _P_U8, ctypes.c_int] + [ctypes.POINTER(ctypes.c_int32)] * 6
    return _dsb


def dsb_encode(vals, nulls, datumlen, version, rle, delta, blocksz=32768):
    """Encode with the REFERENCE writer; returns the framed stream."""
    D = dsb_ref()
    assert D is not None, "reference dsb codec missing"
    v = np.ascontiguousarray(vals, np.int64)
    nl = np.ascontiguousarray(nulls, np.uint8)
    cap = max(1 << 21, len(v) * 16) + blocksz + 64
    out = np.zeros(cap, np.uint8)
    olen = I64()
    nb = ctypes.c_int32()
    rc = D.ref_dsb_encode(v, nl, len(v), datumlen, version, rle, delta,
                          blocksz, out, cap, ctypes.byref(olen),
                          ctypes.byref(nb))
    assert rc == 0, rc
    return out[:olen.value].copy(), nb.value


def dsb_decode(stream, datumlen, version, rle, nmax):
    """Decode with the REFERENCE reader."""
    D = dsb_ref()
    assert D is not None, "reference dsb codec missing"
    stream = np.ascontiguousarray(stream, np.uint8)
    dv = np.zeros(nmax, np.int64)
    dn = np.zeros(nmax, np.uint8)
    n = I64()
    rc = D.ref_dsb_decode(stream, len(stream), datumlen, version, rle,
                          dv, dn, nmax, ctypes.byref(n))
    assert rc == 0, rc
    return dv[:n.value], dn[:n.value]


def ao_wrap(framed, checksums=1, firstrownum=1):
    """Wrap a framed datum-stream into REAL AO storage blocks with the
    REFERENCE's header/checksum writer (cdbappendonlystorageformat.c
    compiled in place)."""
    D = dsb_ref()
    assert D is not None, "reference AO codec missing"
    framed = np.ascontiguousarray(framed, np.uint8)
    cap = len(framed) * 2 + (1 << 16)
    out = np.zeros(cap, np.uint8)
    olen = I64()
    rc = D.ref_ao_wrap_stream(framed, len(framed), checksums, firstrownum,
                              out, cap, ctypes.byref(olen))
    assert rc == 0, rc
    return out[:olen.value].copy()


def ao_wrap_compressed(framed, comptype, complevel, checksums=1,
                       firstrownum=1):
    """Wrap with bulk compression (1=zlib, 2=zstd) via the REFERENCE's
    header writer + the same codec libraries/parameters the reference
    binds (pg_compression.c:253, zstd_compression.c:117)."""
    D = dsb_ref()
    assert D is not None, "reference AO codec missing"
    framed = np.ascontiguousarray(framed, np.uint8)
    cap = len(framed) * 2 + (1 << 16)
    out = np.zeros(cap, np.uint8)
    olen = I64()
    rc = D.ref_ao_wrap_stream_c(framed, len(framed), checksums,
                                firstrownum, comptype, complevel, out,
                                cap, ctypes.byref(olen))
    assert rc == 0, rc
    return out[:olen.value].copy()


def ao_wrap_bulkdense(framed, comptype=0, complevel=0, checksums=1,
                      firstrownum=1):
    """Wrap into BulkDense (long-header) AO blocks via the reference's
    MakeBulkDenseContentHeader — the form used for RLE content with
    bulk compression."""
    D = dsb_ref()
    assert D is not None, "reference AO codec missing"
    framed = np.ascontiguousarray(framed, np.uint8)
    cap = len(framed) * 2 + (1 << 16)
    out = np.zeros(cap, np.uint8)
    olen = I64()
    rc = D.ref_ao_wrap_stream_bd(framed, len(framed), checksums,
                                 firstrownum, comptype, complevel, out,
                                 cap, ctypes.byref(olen))
    assert rc == 0, rc
    return out[:olen.value].copy()


def ao_wrap_large(framed, checksums=1, comptype=0, complevel=0,
                  frag_size=8192):
    """Wrap each frame as a LargeContent metadata block + SmallContent
    fragments (the multi-block form for content larger than one AO
    block), via the reference's MakeLargeContentHeader."""
    D = dsb_ref()
    assert D is not None, "reference AO codec missing"
    framed = np.ascontiguousarray(framed, np.uint8)
    cap = len(framed) * 2 + (1 << 18)
    out = np.zeros(cap, np.uint8)
    olen = I64()
    rc = D.ref_ao_wrap_stream_large(framed, len(framed), checksums,
                                    comptype, complevel, frag_size,
                                    out, cap, ctypes.byref(olen))
    assert rc == 0, rc
    return out[:olen.value].copy()


def ao_probe(block, checksums=1):
    """Parse ONE AO block header with the REFERENCE parser; returns
    (kind, rowcount, datalen, content_off, overall, cksum_ok)."""
    D = dsb_ref()
    assert D is not None, "reference AO codec missing"
    block = np.ascontiguousarray(block, np.uint8)
    outs = [ctypes.c_int32() for _ in range(6)]
    rc = D.ref_ao_probe_block(block, checksums,
                              *[ctypes.byref(o) for o in outs])
    assert rc == 0, rc
    return tuple(o.value for o in outs)


def dsb_encode_text(values, nulls, version, rle, blocksz=32768):
    """Encode python byte-strings with the REFERENCE text writer;
    returns the framed stream."""
    D = dsb_ref()
    assert D is not None, "reference dsb codec missing"
    n = len(values)
    offs = np.zeros(n + 1, np.int64)
    for i, v in enumerate(values):
        offs[i + 1] = offs[i] + (0 if v is None else len(v))
    blob = b"".join(v for v in values if v is not None and len(v))
    # offsets must still be cumulative over ALL values (None -> len 0)
    offs = np.zeros(n + 1, np.int64)
    pos = 0
    parts = []
    for i, v in enumerate(values):
        b = b"" if v is None else v
        parts.append(b)
        pos += len(b)
        offs[i + 1] = pos
    blob = b"".join(parts)
    bts = (np.frombuffer(blob, np.uint8).copy() if blob
           else np.zeros(1, np.uint8))
    nl = np.ascontiguousarray(nulls, np.uint8)
    cap = max(1 << 21, len(blob) * 4 + n * 8) + blocksz + 64
    out = np.zeros(cap, np.uint8)
    olen, nb = I64(), ctypes.c_int32()
    rc = D.ref_dsb_encode_text(bts, offs, nl, n, version, rle, blocksz,
                               out, cap, ctypes.byref(olen),
                               ctypes.byref(nb))
    assert rc == 0, rc
    return out[:olen.value].copy(), nb.value


def dsb_decode_text(stream, version, rle, nmax):
    """Decode text with the REFERENCE reader; returns (values, nulls)
    where values is a list of bytes (b'' for NULL rows)."""
    D = dsb_ref()
    assert D is not None, "reference dsb codec missing"
    stream = np.ascontiguousarray(stream, np.uint8)
    bcap = len(stream) * 2 + 64
    ob = np.zeros(bcap, np.uint8)
    oo = np.zeros(nmax + 1, np.int64)
    on = np.zeros(nmax, np.uint8)
    n = I64()
    rc = D.ref_dsb_decode_text(stream, len(stream), version, rle, ob,
                               bcap, oo, on, nmax, ctypes.byref(n))
    assert rc == 0, rc
    vals = [bytes(ob[oo[i]:oo[i + 1]]) for i in range(n.value)]
    return vals, on[:n.value].copy()


class MtSchema:
    """A fixed-width memtuple schema bound through the REFERENCE's
    create_memtuple_binding (memtuple.c compiled in place)."""

    def __init__(self, attlen, attalign, atttypid=None):
        D = dsb_ref()
        assert D is not None, "reference memtuple codec missing"
        self.D = D
        self.natts = len(attlen)
        self.attlen = list(attlen)
        if atttypid is None:
            atttypid = [{8: 20, 4: 23, 2: 21, 1: 18, -1: 25}[l]
                        for l in attlen]
        al = (ctypes.c_int32 * self.natts)(*attlen)
        bv = np.array([0 if l == -1 else 1 for l in attlen], np.uint8)
        ti = (ctypes.c_uint32 * self.natts)(*atttypid)
        self.bind = D.ref_mt_create_binding(
            self.natts, al, bv, "".join(attalign).encode(), ti)
        assert self.bind

    def form(self, values, isnull):
        v = np.array([int(x) for x in values], np.int64)
        nl = np.array([int(x) for x in isnull], np.uint8)
        out = np.zeros(1 << 16, np.uint8)
        ln = self.D.ref_mt_form(self.bind, v, nl, out, 1 << 16)
        assert ln > 0, ln
        return out[:ln].copy()

    def getattr(self, tup, attnum):
        tup = np.ascontiguousarray(tup, np.uint8)
        v, isn = I64(), ctypes.c_uint8()
        rc = self.D.ref_mt_getattr(self.bind, tup, attnum,
                                   ctypes.byref(v), ctypes.byref(isn))
        assert rc == 0
        return v.value, isn.value

    def form_var(self, values, isnull):
        """Form a tuple with mixed fixed/text attrs: text attrs take a
        python bytes value."""
        is_text = np.array([1 if l == -1 else 0 for l in self.attlen],
                           np.uint8)
        blob_parts, offs, ints = [], [0], []
        pos = 0
        for l, v in zip(self.attlen, values):
            if l == -1:
                b = v if isinstance(v, (bytes, bytearray)) else b""
                blob_parts.append(bytes(b))
                ints.append(len(offs) - 1)
                pos += len(b)
                offs.append(pos)
            else:
                ints.append(int(v))
        blob = b"".join(blob_parts)
        bts = (np.frombuffer(blob, np.uint8).copy() if blob
               else np.zeros(1, np.uint8))
        offs_a = np.array(offs, np.int64)
        v = np.array(ints, np.int64)
        nl = np.array([int(x) for x in isnull], np.uint8)
        out = np.zeros(1 << 20, np.uint8)
        ln = self.D.ref_mt_form_var(self.bind, v, is_text, bts, offs_a,
                                    nl, out, 1 << 20)
        assert ln > 0, ln
        return out[:ln].copy()

    def colbind(self, attnum, large=False):
        outs = [ctypes.c_int32() for _ in range(6)]
        rc = self.D.ref_mt_get_colbind(self.bind, attnum, int(large),
                                       *[ctypes.byref(o) for o in outs])
        assert rc == 0
        return tuple(o.value for o in outs)

    def info(self):
        outs = [ctypes.c_int32() for _ in range(3)]
        self.D.ref_mt_binding_info(self.bind,
                                   *[ctypes.byref(o) for o in outs])
        return tuple(o.value for o in outs)


def pgdate(y, m, d):
    return lib().gg_oracle_pgdate(y, m, d)


def numeric_str(val128, scale):
    buf = ctypes.create_string_buffer(80)
    lo = val128 & ((1 << 64) - 1)
    hi = val128 >> 64
    lib().gg_numeric_to_str(U64(lo), I64(hi), scale, buf)
    return buf.value.decode()


def avg_str(sum128, sum_scale, count):
    buf = ctypes.create_string_buffer(80)
    lo = sum128 & ((1 << 64) - 1)
    hi = sum128 >> 64
    lib().gg_numeric_avg_to_str(U64(lo), I64(hi), sum_scale, I64(count), buf)
    return buf.value.decode()


def q1_arrays(shipdate, rflag, lstatus, qty, price, disc, tax, cutoff,
              nthreads=0):
    out = (Q1Group * 6)()
    rc = lib().gg_oracle_q1_arrays(
        shipdate, rflag, lstatus, qty, price, disc, tax,
        len(shipdate), cutoff, out, nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q1_synth(seed, sf, cutoff, row_lo=0, row_hi=-1, nthreads=0):
    out = (Q1Group * 6)()
    rc = lib().gg_oracle_q1_synth(seed, sf, row_lo, row_hi, cutoff, out,
                                  nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q1_volcano_synth(seed, sf, cutoff, row_lo=0, row_hi=-1, nthreads=1):
    """Tuple-at-a-time Volcano restatement (BASELINE.md cpu-ref leg)."""
    out = (Q1Group * 6)()
    rc = lib().gg_oracle_q1_volcano_synth(seed, sf, row_lo, row_hi, cutoff,
                                          out, nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q1_synth_segment(seed, sf, nseg, seg, cutoff, nthreads=0):
    out = (Q1Group * 6)()
    rc = lib().gg_oracle_q1_synth_segment(seed, sf, nseg, seg, cutoff, out,
                                          nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q1_finalize(groups):
    """Occupied groups in (returnflag, linestatus) order, with the
    reference's output strings (bb_mpph mpph1 column order)."""
    names = [("A", "F"), ("A", "O"), ("N", "F"), ("N", "O"), ("R", "F"),
             ("R", "O")]
    rows = []
    for g, (rf, ls) in zip(groups, names):
        if g["count"] == 0:
            continue
        rows.append({
            "l_returnflag": rf,
            "l_linestatus": ls,
            "sum_qty": numeric_str(g["sum_qty_c"], 2),
            "sum_base_price": numeric_str(g["sum_base_c"], 2),
            "sum_disc_price": numeric_str(g["sum_disc4"], 4),
            "sum_charge": numeric_str(g["sum_charge6"], 6),
            "avg_qty": avg_str(g["sum_qty_c"], 2, g["count"]),
            "avg_price": avg_str(g["sum_base_c"], 2, g["count"]),
            "avg_disc": avg_str(g["sum_dcol_c"], 2, g["count"]),
            "count_order": g["count"],
        })
    return rows


def q3_arrays(c_custkey, c_mktseg, seg_code, o_orderkey, o_custkey,
              o_orderdate, o_shippriority, l_orderkey, l_shipdate, l_price,
              l_disc, cutoff, k=10, nthreads=0):
    topk = (Q3Row * 64)()
    res = Q3Result()
    rc = lib().gg_oracle_q3_arrays(
        c_custkey, c_mktseg, len(c_custkey), seg_code,
        o_orderkey, o_custkey, o_orderdate, o_shippriority, len(o_orderkey),
        l_orderkey, l_shipdate, l_price, l_disc, len(l_orderkey),
        cutoff, k, topk, ctypes.byref(res), nthreads)
    assert rc == 0, rc
    return ([topk[i].as_dict() for i in range(res.n_out)], res.as_dict())


def q3_synth(seed, sf, cutoff, k=10, nthreads=0):
    topk = (Q3Row * 64)()
    res = Q3Result()
    rc = lib().gg_oracle_q3_synth(seed, sf, cutoff, k, topk,
                                  ctypes.byref(res), nthreads)
    assert rc == 0, rc
    return ([topk[i].as_dict() for i in range(res.n_out)], res.as_dict())


def sumprice_synth(seed, sf, cutoff, nthreads=0):
    s = I64()
    c = I64()
    rc = lib().gg_oracle_sumprice_synth(seed, sf, cutoff, ctypes.byref(s),
                                        ctypes.byref(c), nthreads)
    assert rc == 0, rc
    return s.value, c.value


def sumprice_arrays(shipdate, price, cutoff, nthreads=0):
    s = I64()
    c = I64()
    rc = lib().gg_oracle_sumprice_arrays(shipdate, price, len(shipdate),
                                         cutoff, ctypes.byref(s),
                                         ctypes.byref(c), nthreads)
    assert rc == 0, rc
    return s.value, c.value


def gen_lineitem(seed, row_lo, row_hi):
    n = row_hi - row_lo
    cols = dict(
        orderkey=np.empty(n, np.int64), qty=np.empty(n, np.int64),
        price=np.empty(n, np.int64), disc=np.empty(n, np.int64),
        tax=np.empty(n, np.int64), shipdate=np.empty(n, np.int32),
        rflag=np.empty(n, np.uint8), lstatus=np.empty(n, np.uint8))
    lib().gg_oracle_gen_lineitem(
        seed, row_lo, row_hi, cols["orderkey"], cols["qty"], cols["price"],
        cols["disc"], cols["tax"], cols["shipdate"], cols["rflag"],
        cols["lstatus"])
    return cols


def gen_orders(seed, sf, row_lo, row_hi):
    n = row_hi - row_lo
    cols = dict(
        orderkey=np.empty(n, np.int64), custkey=np.empty(n, np.int64),
        orderdate=np.empty(n, np.int32), shippriority=np.empty(n, np.int32))
    lib().gg_oracle_gen_orders(seed, sf, row_lo, row_hi, cols["orderkey"],
                               cols["custkey"], cols["orderdate"],
                               cols["shippriority"])
    return cols


def gen_customer(seed, row_lo, row_hi):
    n = row_hi - row_lo
    cols = dict(custkey=np.empty(n, np.int64), mktseg=np.empty(n, np.uint8),
                nationkey=np.empty(n, np.uint8))
    lib().gg_oracle_gen_customer(seed, row_lo, row_hi, cols["custkey"],
                                 cols["mktseg"], cols["nationkey"])
    return cols


def gen_supplier(seed, row_lo, row_hi):
    n = row_hi - row_lo
    cols = dict(suppkey=np.empty(n, np.int64),
                nationkey=np.empty(n, np.uint8))
    lib().gg_oracle_gen_supplier(seed, row_lo, row_hi, cols["suppkey"],
                                 cols["nationkey"])
    return cols


def gen_l_suppkey(seed, sf, row_lo, row_hi):
    out = np.empty(row_hi - row_lo, np.int64)
    lib().gg_oracle_gen_l_suppkey(seed, sf, row_lo, row_hi, out)
    return out


# nation names in nationkey order (reference fixture nation.csv)
NATION_NAMES = [
    "ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
    "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
    "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA", "ROMANIA",
    "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM", "UNITED STATES"]
NATION_REGION = [0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0,
                 0, 0, 1, 2, 3, 4, 2, 3, 3, 1]
REGION_NAMES = ["AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST"]


def q5_synth(seed, sf, regionkey, date_lo, date_hi, nthreads=0):
    out = (Q5Group * 25)()
    rc = lib().gg_oracle_q5_synth(seed, sf, regionkey, date_lo, date_hi,
                                  out, nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q5_arrays(c_custkey, c_nation, o_orderkey, o_custkey, o_orderdate,
              l_orderkey, l_suppkey, l_price, l_disc, s_suppkey, s_nation,
              regionkey, date_lo, date_hi, nation_region=None, nthreads=0):
    out = (Q5Group * 25)()
    nr = np.array(nation_region if nation_region is not None
                  else NATION_REGION, np.int32)
    rc = lib().gg_oracle_q5_arrays(
        c_custkey, c_nation, len(c_custkey),
        o_orderkey, o_custkey, o_orderdate, len(o_orderkey),
        l_orderkey, l_suppkey, l_price, l_disc, len(l_orderkey),
        s_suppkey, s_nation, len(s_suppkey),
        nr, regionkey, date_lo, date_hi, out, nthreads)
    assert rc == 0, rc
    return [g.as_dict() for g in out]


def q5_rows(groups):
    """Occupied groups as output rows ordered by revenue desc (the
    query's ORDER BY; ties refined by n_name asc — both sides use it)."""
    rows = [{"nationkey": n, "n_name": NATION_NAMES[n],
             "revenue4": g["revenue4"], "count": g["count"]}
            for n, g in enumerate(groups) if g["count"]]
    rows.sort(key=lambda r: (-r["revenue4"], r["n_name"]))
    return rows

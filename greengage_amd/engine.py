"""ctypes bindings for the C-ABI engine (include/engine_abi.h).

The engine library is product native code (HIP gfx950 + RCCL); this
module is plumbing only.  It fails loudly if the library is missing —
no silent CPU fallback (project rule: GPU tests must run the native
path).
"""
import ctypes
import datetime
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "libgreengage_engine.so")

I8 = ctypes.c_uint8
I32 = ctypes.c_int32
I64 = ctypes.c_int64
U64 = ctypes.c_uint64


class EngineError(RuntimeError):
    pass


class _Config(ctypes.Structure):
    _fields_ = [("device", ctypes.c_int), ("n_segments", ctypes.c_int),
                ("segment_id", ctypes.c_int), ("hbm_limit_bytes", U64)]


class _ColumnDesc(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char_p), ("type", ctypes.c_int),
                ("host_data", ctypes.c_void_p),
                ("device_data", ctypes.c_void_p)]


class _PipelineDesc(ctypes.Structure):
    _fields_ = [("kind", ctypes.c_int), ("lineitem", I32), ("orders", I32),
                ("customer", I32), ("supplier", I32), ("nation", I32),
                ("cutoff_date", I32), ("cutoff_hi", I32), ("mktsegment", I8),
                ("regionkey", I8), ("limit_k", I64)]


class _PlanPred(ctypes.Structure):
    _fields_ = [("col", ctypes.c_char_p), ("lo", I64), ("hi", I64)]


class _PlanScan(ctypes.Structure):
    _fields_ = [("table", I32), ("preds", _PlanPred * 8),
                ("npreds", ctypes.c_int)]


class _PlanJoin(ctypes.Structure):
    _fields_ = [("build", _PlanScan), ("build_key", ctypes.c_char_p),
                ("probe_key", ctypes.c_char_p)]


class _PlanAgg(ctypes.Structure):
    _fields_ = [("kind", ctypes.c_int), ("nfactors", ctypes.c_int),
                ("col", ctypes.c_char_p * 3), ("mod", ctypes.c_int8 * 3)]


class _PlanDesc(ctypes.Structure):
    _fields_ = [("scan", _PlanScan), ("joins", _PlanJoin * 2),
                ("njoins", ctypes.c_int),
                ("group_cols", ctypes.c_char_p * 2),
                ("ngroup", ctypes.c_int), ("aggs", _PlanAgg * 8),
                ("naggs", ctypes.c_int)]


NEG_INF = -(1 << 63)          # one-sided range bounds
POS_INF = (1 << 63) - 1
PLAN_NULL_KEY = -(1 << 63) + 1


class KernelStat(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char * 48), ("launches", I64),
                ("total_ms", ctypes.c_double), ("rows_in", I64),
                ("rows_out", I64), ("hbm_bytes_algorithmic", I64)]


COLTYPE = {"int64": 0, "int32": 1, "dec64": 2, "char1": 3}
PIPE_Q1, PIPE_Q3, PIPE_SUMPRICE, PIPE_Q5 = 1, 2, 3, 4

# n_name strings in nationkey order (reference fixture nation.csv)
NATION_NAMES = [
    "ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
    "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
    "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA", "ROMANIA",
    "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM", "UNITED STATES"]

_EPOCH = datetime.date(2000, 1, 1)


def pgdate(y, m, d):
    """PG DateADT encoding (int32 days since 2000-01-01)."""
    return (datetime.date(y, m, d) - _EPOCH).days


def PGDate(iso):
    y, m, d = map(int, iso.split("-"))
    return pgdate(y, m, d)


def build_library():
    subprocess.run(["make", "-s", "-C", os.path.join(_DIR, "csrc")],
                   check=True)


def _load():
    if not os.path.exists(_LIB_PATH):
        raise EngineError(
            f"engine library missing: {_LIB_PATH}. Build it with "
            "`make -C greengage_amd/csrc` (or __graft_entry__.build()). "
            "There is no CPU fallback.")
    lib = ctypes.CDLL(_LIB_PATH)
    lib.gg_engine_last_error.restype = ctypes.c_char_p
    lib.gg_engine_build_info.restype = ctypes.c_char_p
    lib.gg_engine_init.argtypes = [ctypes.POINTER(_Config)]
    lib.gg_engine_register_table.argtypes = [
        ctypes.c_char_p, ctypes.POINTER(_ColumnDesc), ctypes.c_int, I64,
        ctypes.POINTER(I32)]
    lib.gg_engine_register_synth.argtypes = [ctypes.c_char_p, U64, I64,
                                             ctypes.POINTER(I32)]
    lib.gg_engine_table_nrows.argtypes = [I32, ctypes.POINTER(I64)]
    lib.gg_engine_fetch_column.argtypes = [I32, ctypes.c_char_p,
                                           ctypes.c_void_p, ctypes.c_size_t]
    lib.gg_engine_compile_pipeline.argtypes = [
        ctypes.POINTER(_PipelineDesc), ctypes.POINTER(I32)]
    lib.gg_engine_execute.argtypes = [I32, ctypes.c_void_p, ctypes.c_size_t,
                                      ctypes.POINTER(ctypes.c_size_t)]
    lib.gg_engine_stats.argtypes = [I32, ctypes.POINTER(KernelStat),
                                    ctypes.c_int, ctypes.POINTER(ctypes.c_int)]
    lib.gg_engine_comm_id.argtypes = [ctypes.c_void_p]
    lib.gg_engine_comm_init.argtypes = [ctypes.c_void_p]
    lib.gg_engine_compile_plan.argtypes = [ctypes.POINTER(_PlanDesc),
                                           ctypes.POINTER(I32)]
    lib.gg_engine_table_set_nulls.argtypes = [I32, ctypes.c_char_p,
                                              ctypes.c_void_p]
    lib.gg_engine_hash_groupby_i64_n.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        I64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, I64,
        ctypes.POINTER(I64)]
    lib.gg_engine_numeric_str.argtypes = [U64, I64, ctypes.c_int,
                                          ctypes.c_char_p]
    lib.gg_engine_radix_sort_u64.argtypes = [ctypes.c_void_p,
                                             ctypes.c_void_p, I64,
                                             ctypes.c_int, ctypes.c_int]
    lib.gg_engine_hash_groupby_i64.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, I64, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_void_p, I64, ctypes.POINTER(I64)]
    lib.gg_engine_aocs_decode.argtypes = [
        ctypes.c_void_p, I64, ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_int, ctypes.c_void_p, I64, ctypes.POINTER(I64)]
    lib.gg_engine_aocs_decode_text.restype = ctypes.c_int
    lib.gg_engine_aocs_decode_text.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.c_void_p, ctypes.c_int64, ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_aocs_decode_ao_text.restype = ctypes.c_int
    lib.gg_engine_aocs_decode_ao_text.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int, ctypes.c_int,
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p,
        ctypes.c_int64, ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_hash_join_i64_spill.restype = ctypes.c_int
    lib.gg_engine_hash_join_i64_spill.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_hash_groupby_i64_spill.restype = ctypes.c_int
    lib.gg_engine_hash_groupby_i64_spill.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32)]
    class _AOCol(ctypes.Structure):
        _fields_ = [("name", ctypes.c_char_p),
                    ("type", ctypes.c_int),
                    ("stream", ctypes.c_void_p),
                    ("stream_len", ctypes.c_int64),
                    ("checksums", ctypes.c_int),
                    ("ao_version", ctypes.c_int),
                    ("dsb_version", ctypes.c_int),
                    ("comptype", ctypes.c_int),
                    ("text_dict", ctypes.c_int),
                    ("nullable", ctypes.c_int)]
    lib.gg_AOCol = _AOCol
    lib.gg_engine_table_text_dict.restype = ctypes.c_int
    lib.gg_engine_table_text_dict.argtypes = [
        ctypes.c_int32, ctypes.c_char_p, ctypes.c_void_p,
        ctypes.c_int64, ctypes.c_void_p, ctypes.c_int32,
        ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_register_table_ao.restype = ctypes.c_int
    lib.gg_engine_register_table_ao.argtypes = [
        ctypes.c_char_p, ctypes.POINTER(_AOCol), ctypes.c_int,
        ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_text_dict_encode.restype = ctypes.c_int
    lib.gg_engine_text_dict_encode.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int32,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_motion_chunkify.restype = ctypes.c_int
    lib.gg_engine_motion_chunkify.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int32, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int64, ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_motion_dechunkify.restype = ctypes.c_int
    lib.gg_engine_motion_dechunkify.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p,
        ctypes.c_int64, ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_int)]
    lib.gg_engine_memtuple_binding.restype = ctypes.c_int
    lib.gg_engine_memtuple_binding.argtypes = [
        ctypes.c_int, ctypes.POINTER(ctypes.c_int32), ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_memtuple_binding_large.restype = ctypes.c_int
    lib.gg_engine_memtuple_binding_large.argtypes = [
        ctypes.c_int, ctypes.POINTER(ctypes.c_int32), ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_int32)]
    lib.gg_engine_memtuple_encode.restype = ctypes.c_int
    lib.gg_engine_memtuple_encode.argtypes = [
        ctypes.c_int, ctypes.POINTER(ctypes.c_int32), ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_void_p),
        ctypes.c_int64, ctypes.c_void_p, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_memtuple_decode.restype = ctypes.c_int
    lib.gg_engine_memtuple_decode.argtypes = [
        ctypes.c_int, ctypes.POINTER(ctypes.c_int32), ctypes.c_char_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.POINTER(ctypes.c_void_p),
        ctypes.POINTER(ctypes.c_void_p), ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_aocs_decode_ao.restype = ctypes.c_int
    lib.gg_engine_aocs_decode_ao.argtypes = [
        ctypes.POINTER(ctypes.c_uint8), ctypes.c_int64, ctypes.c_int,
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int,
        ctypes.POINTER(ctypes.c_uint8), ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64)]
    lib.gg_engine_avg_str.argtypes = [U64, I64, ctypes.c_int, I64,
                                      ctypes.c_char_p]
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


def _check(rc, what):
    if rc != 0:
        raise EngineError(
            f"{what}: status {rc}: {lib().gg_engine_last_error().decode()}")


def _i128(lo, hi):
    return (hi << 64) | lo


class Engine:
    """One engine instance per process (= per GPU = per segment)."""

    def __init__(self, device=0, n_segments=1, segment_id=0):
        cfg = _Config(device=device, n_segments=n_segments,
                      segment_id=segment_id, hbm_limit_bytes=0)
        _check(lib().gg_engine_init(ctypes.byref(cfg)), "engine_init")
        self.n_segments = n_segments
        self.segment_id = segment_id

    def shutdown(self):
        _check(lib().gg_engine_shutdown(), "engine_shutdown")

    # ---- comm (Motion-equivalent RCCL bootstrap) ----
    def comm_id(self):
        buf = ctypes.create_string_buffer(128)
        _check(lib().gg_engine_comm_id(buf), "comm_id")
        return buf.raw

    def comm_init(self, id_bytes):
        buf = ctypes.create_string_buffer(bytes(id_bytes), 128)
        _check(lib().gg_engine_comm_init(buf), "comm_init")

    # ---- tables ----
    def register_synth(self, name, seed=42, sf=1):
        h = I32()
        _check(lib().gg_engine_register_synth(name.encode(), seed, sf,
                                              ctypes.byref(h)),
               f"register_synth({name})")
        return h.value

    def register_table(self, name, cols, nrows):
        """cols: list of (name, typename, numpy_array)."""
        descs = (_ColumnDesc * len(cols))()
        keep = []
        for i, (cname, ctype, arr) in enumerate(cols):
            arr = _ascontig(arr, ctype)
            keep.append(arr)
            descs[i].name = cname.encode()
            descs[i].type = COLTYPE[ctype]
            descs[i].host_data = arr.ctypes.data_as(ctypes.c_void_p)
            descs[i].device_data = None
        h = I32()
        _check(lib().gg_engine_register_table(name.encode(), descs,
                                              len(cols), nrows,
                                              ctypes.byref(h)),
               f"register_table({name})")
        return h.value

    def register_table_ao(self, name, cols):
        """Mount a real AO table: cols = list of (name, typename,
        ao_bytes, checksums, ao_version, dsb_version, comptype)."""
        import numpy as np
        L = lib()
        descs = (L.gg_AOCol * len(cols))()
        keep = []
        for i, col in enumerate(cols):
            cname, ctype, ao, cks, aov, dsbv, ct = col[:7]
            ao = np.ascontiguousarray(ao, np.uint8)
            keep.append(ao)
            descs[i].name = cname.encode()
            descs[i].type = COLTYPE[ctype]
            descs[i].stream = ao.ctypes.data_as(ctypes.c_void_p)
            descs[i].stream_len = len(ao)
            descs[i].checksums = cks
            descs[i].ao_version = aov
            descs[i].dsb_version = dsbv
            descs[i].comptype = ct
            descs[i].text_dict = col[7] if len(col) > 7 else 0
            descs[i].nullable = col[8] if len(col) > 8 else 0
        h = I32()
        _check(L.gg_engine_register_table_ao(name.encode(), descs,
                                             len(cols),
                                             ctypes.byref(h)),
               f"register_table_ao({name})")
        return h.value

    def table_text_dict(self, h, col):
        """Sorted dictionary of a text_dict-mounted column."""
        import numpy as np
        out = np.zeros(1 << 16, np.uint8)
        offs = np.zeros(257, np.int64)
        n = ctypes.c_int32()
        _check(lib().gg_engine_table_text_dict(
            h, col.encode(), out.ctypes.data_as(ctypes.c_void_p),
            len(out), offs.ctypes.data_as(ctypes.c_void_p), 256,
            ctypes.byref(n)), "table_text_dict")
        return [bytes(out[offs[i]:offs[i + 1]]) for i in range(n.value)]

    def table_nrows(self, h):
        n = I64()
        _check(lib().gg_engine_table_nrows(h, ctypes.byref(n)), "nrows")
        return n.value

    def fetch_column(self, h, col_name, dtype):
        import numpy as np
        n = self.table_nrows(h)
        arr = np.empty(n, dtype=dtype)
        _check(lib().gg_engine_fetch_column(
            h, col_name.encode(), arr.ctypes.data_as(ctypes.c_void_p),
            arr.nbytes), f"fetch_column({col_name})")
        return arr

    # ---- pipelines ----
    def compile(self, kind, lineitem=-1, orders=-1, customer=-1,
                supplier=-1, nation=-1, cutoff_date=0, cutoff_hi=0,
                mktsegment=0, regionkey=0, limit_k=10):
        d = _PipelineDesc(kind=kind, lineitem=lineitem, orders=orders,
                          customer=customer, supplier=supplier,
                          nation=nation, cutoff_date=cutoff_date,
                          cutoff_hi=cutoff_hi, mktsegment=mktsegment,
                          regionkey=regionkey, limit_k=limit_k)
        h = I32()
        _check(lib().gg_engine_compile_pipeline(ctypes.byref(d),
                                                ctypes.byref(h)), "compile")
        return h.value

    # ---- generalized pipeline descriptor (v2) ----
    def compile_plan(self, scan_table, preds=(), joins=(), group_cols=(),
                     aggs=()):
        """Compile a compositional plan (engine_abi.h gg_plan_desc).

        preds: [(col, lo, hi)]  (half-open; NEG_INF/POS_INF one-sided)
        joins: [{"table": h, "build_key": c, "probe_key": c,
                 "preds": [...]}]  (hash SEMI-join filters)
        group_cols: 0-2 column names of the driving table
        aggs: ["count"] | ("count", col) |
              ("sum", [(col, "id"|"sub100"|"add100"), ...])
        """
        mods = {"id": 0, "sub100": 1, "add100": 2}

        def fill_scan(dst, table, plist):
            dst.table = table
            dst.npreds = len(plist)
            for i, (c, lo, hi) in enumerate(plist):
                dst.preds[i].col = c.encode()
                dst.preds[i].lo = lo
                dst.preds[i].hi = hi

        d = _PlanDesc()
        fill_scan(d.scan, scan_table, list(preds))
        d.njoins = len(joins)
        for j, spec in enumerate(joins):
            fill_scan(d.joins[j].build, spec["table"],
                      list(spec.get("preds", ())))
            d.joins[j].build_key = spec["build_key"].encode()
            d.joins[j].probe_key = spec["probe_key"].encode()
        d.ngroup = len(group_cols)
        for g, c in enumerate(group_cols):
            d.group_cols[g] = c.encode()
        d.naggs = len(aggs)
        for a, spec in enumerate(aggs):
            if spec == "count" or spec == ("count",):
                d.aggs[a].kind = 0
                d.aggs[a].nfactors = 0
            elif spec[0] == "count":
                d.aggs[a].kind = 1
                d.aggs[a].nfactors = 1
                d.aggs[a].col[0] = spec[1].encode()
                d.aggs[a].mod[0] = 0
            else:
                assert spec[0] == "sum", spec
                d.aggs[a].kind = 2
                d.aggs[a].nfactors = len(spec[1])
                for f, (c, m) in enumerate(spec[1]):
                    d.aggs[a].col[f] = c.encode()
                    d.aggs[a].mod[f] = mods[m]
        h = I32()
        _check(lib().gg_engine_compile_plan(ctypes.byref(d),
                                            ctypes.byref(h)), "compile_plan")
        return h.value

    def execute_plan(self, p, max_groups=1 << 16):
        naggs_guess = 8
        raw = self.execute_raw(
            p, 16 + max_groups * (16 + 16 * naggs_guess))
        ng = int.from_bytes(raw[0:8], "little", signed=True)
        naggs = int.from_bytes(raw[8:12], "little", signed=True)
        groups = []
        off = 16
        for _ in range(ng):
            k0 = int.from_bytes(raw[off:off + 8], "little", signed=True)
            k1 = int.from_bytes(raw[off + 8:off + 16], "little",
                                signed=True)
            off += 16
            vals = []
            for _ in range(naggs):
                lo = int.from_bytes(raw[off:off + 8], "little")
                hi = int.from_bytes(raw[off + 8:off + 16], "little",
                                    signed=True)
                vals.append((hi << 64) | lo)
                off += 16
            groups.append((k0, k1, vals))
        return groups

    def set_nulls(self, table, col, nulls):
        import numpy as np
        nl = np.ascontiguousarray(nulls, np.uint8)
        _check(lib().gg_engine_table_set_nulls(
            table, col.encode(), nl.ctypes.data_as(ctypes.c_void_p)),
            "set_nulls")

    @staticmethod
    def hash_groupby_n(keys, key_nulls, vals, val_nulls):
        import numpy as np
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.int64)
        kn = (None if key_nulls is None
              else np.ascontiguousarray(key_nulls, np.uint8))
        vn = (None if val_nulls is None
              else np.ascontiguousarray(val_nulls, np.uint8))
        cap = len(keys) if len(keys) else 1
        ok = np.empty(cap, np.int64)
        os_ = np.empty(cap, np.int64)
        oc = np.empty(cap, np.int64)
        ng = I64()
        _check(lib().gg_engine_hash_groupby_i64_n(
            keys.ctypes.data_as(ctypes.c_void_p),
            None if kn is None else kn.ctypes.data_as(ctypes.c_void_p),
            vals.ctypes.data_as(ctypes.c_void_p),
            None if vn is None else vn.ctypes.data_as(ctypes.c_void_p),
            len(keys), ok.ctypes.data_as(ctypes.c_void_p),
            os_.ctypes.data_as(ctypes.c_void_p),
            oc.ctypes.data_as(ctypes.c_void_p), cap,
            ctypes.byref(ng)), "hash_groupby_n")
        n = ng.value
        return ok[:n], os_[:n], oc[:n]

    def execute_raw(self, p, arena_bytes):
        arena = ctypes.create_string_buffer(arena_bytes)
        written = ctypes.c_size_t()
        _check(lib().gg_engine_execute(p, arena, arena_bytes,
                                       ctypes.byref(written)), "execute")
        return arena.raw[:written.value]

    def stats(self, p):
        out = (KernelStat * 32)()
        n = ctypes.c_int()
        _check(lib().gg_engine_stats(p, out, 32, ctypes.byref(n)), "stats")
        return [{
            "name": out[i].name.decode(),
            "launches": out[i].launches,
            "total_ms": out[i].total_ms,
            "rows_in": out[i].rows_in,
            "rows_out": out[i].rows_out,
            "hbm_bytes_algorithmic": out[i].hbm_bytes_algorithmic,
        } for i in range(n.value)]

    # ---- typed result decoding (gg_result.h layouts) ----
    def execute_q1(self, p):
        raw = self.execute_raw(p, 8 + 6 * 72)
        n_groups = int.from_bytes(raw[0:4], "little", signed=True)
        groups = []
        off = 8
        for _ in range(6):
            f = _unpack_q1_group(raw[off:off + 72])
            off += 72
            if f["count"]:
                groups.append(f)
        assert len(groups) == n_groups
        return groups

    def execute_q3(self, p, k=10):
        raw = self.execute_raw(p, 48 + 32 * max(k, 1))
        hdr = {
            "n_out": int.from_bytes(raw[0:8], "little", signed=True),
            "n_groups": int.from_bytes(raw[8:16], "little", signed=True),
            "rev_sum4": _i128(int.from_bytes(raw[16:24], "little"),
                              int.from_bytes(raw[24:32], "little",
                                             signed=True)),
            "group_checksum": int.from_bytes(raw[32:40], "little"),
            "n_join_rows": int.from_bytes(raw[40:48], "little", signed=True),
        }
        rows = []
        off = 48
        for _ in range(hdr["n_out"]):
            b = raw[off:off + 32]
            off += 32
            rows.append({
                "orderkey": int.from_bytes(b[0:8], "little", signed=True),
                "revenue4": _i128(int.from_bytes(b[8:16], "little"),
                                  int.from_bytes(b[16:24], "little",
                                                 signed=True)),
                "orderdate": int.from_bytes(b[24:28], "little", signed=True),
                "shippriority": int.from_bytes(b[28:32], "little",
                                               signed=True),
            })
        return rows, hdr

    def execute_q5(self, p):
        raw = self.execute_raw(p, 8 + 25 * 32)
        n_out = int.from_bytes(raw[0:8], "little", signed=True)
        rows = []
        off = 8
        for _ in range(n_out):
            b = raw[off:off + 32]
            off += 32
            nk = int.from_bytes(b[0:4], "little", signed=True)
            rows.append({
                "nationkey": nk,
                "n_name": NATION_NAMES[nk],
                "count": int.from_bytes(b[8:16], "little", signed=True),
                "revenue4": _i128(int.from_bytes(b[16:24], "little"),
                                  int.from_bytes(b[24:32], "little",
                                                 signed=True)),
            })
        return rows

    def execute_sumprice(self, p):
        raw = self.execute_raw(p, 16)
        return (int.from_bytes(raw[0:8], "little", signed=True),
                int.from_bytes(raw[8:16], "little", signed=True))

    # ---- AOCS datum-stream decode (on-disk columnar blocks) ----
    @staticmethod
    def aocs_decode(stream, version, datumlen, nmax, out_width=8):
        import numpy as np
        stream = np.ascontiguousarray(stream, dtype=np.uint8)
        vals = np.empty(nmax, np.int32 if out_width == 4 else np.int64)
        nulls = np.empty(nmax, np.uint8)
        n = I64()
        _check(lib().gg_engine_aocs_decode(
            stream.ctypes.data_as(ctypes.c_void_p), len(stream), version,
            datumlen, vals.ctypes.data_as(ctypes.c_void_p), out_width,
            nulls.ctypes.data_as(ctypes.c_void_p), nmax,
            ctypes.byref(n)), "aocs_decode")
        return vals[:n.value], nulls[:n.value]

    @staticmethod
    def aocs_decode_ao(stream, checksums, ao_version, dsb_version,
                       datumlen, nmax, out_width=8, comptype=0):
        """Decode REAL AO storage blocks: restated header parse +
        CRC32C verify on the host, datum-stream content on the GPU."""
        import numpy as np
        stream = np.ascontiguousarray(stream, dtype=np.uint8)
        vals = np.empty(nmax, np.int32 if out_width == 4 else np.int64)
        nulls = np.empty(nmax, np.uint8)
        n = I64()
        _check(lib().gg_engine_aocs_decode_ao(
            stream.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            len(stream), checksums, ao_version, dsb_version, comptype,
            datumlen,
            vals.ctypes.data_as(ctypes.c_void_p), out_width,
            nulls.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)), nmax,
            ctypes.byref(n)), "aocs_decode_ao")
        return vals[:n.value], nulls[:n.value]

    @staticmethod
    def aocs_decode_text(stream, version, nmax):
        """GPU decode of TEXT datum-stream blocks; returns
        (list of bytes, nulls)."""
        import numpy as np
        stream = np.ascontiguousarray(stream, np.uint8)
        offs = np.empty(nmax, np.uint64)
        lens = np.empty(nmax, np.uint32)
        nulls = np.empty(nmax, np.uint8)
        pool = np.empty(len(stream) + 64, np.uint8)
        n, plen = I64(), I64()
        _check(lib().gg_engine_aocs_decode_text(
            stream.ctypes.data_as(ctypes.c_void_p), len(stream), version,
            offs.ctypes.data_as(ctypes.c_void_p),
            lens.ctypes.data_as(ctypes.c_void_p),
            nulls.ctypes.data_as(ctypes.c_void_p), nmax,
            pool.ctypes.data_as(ctypes.c_void_p), len(pool),
            ctypes.byref(n), ctypes.byref(plen)), "aocs_decode_text")
        vals = [bytes(pool[offs[i]:offs[i] + lens[i]])
                for i in range(n.value)]
        return vals, nulls[:n.value].copy()

    @staticmethod
    def aocs_decode_ao_text(stream, checksums, ao_version, dsb_version,
                            nmax, comptype=0):
        """Full AO read path for TEXT columns: header/CRC/codec layer
        on the host, GPU varlena decode."""
        import numpy as np
        stream = np.ascontiguousarray(stream, np.uint8)
        offs = np.empty(nmax, np.uint64)
        lens = np.empty(nmax, np.uint32)
        nulls = np.empty(nmax, np.uint8)
        pool = np.empty(len(stream) * 8 + (1 << 16), np.uint8)
        n, plen = I64(), I64()
        _check(lib().gg_engine_aocs_decode_ao_text(
            stream.ctypes.data_as(ctypes.c_void_p), len(stream),
            checksums, ao_version, dsb_version, comptype,
            offs.ctypes.data_as(ctypes.c_void_p),
            lens.ctypes.data_as(ctypes.c_void_p),
            nulls.ctypes.data_as(ctypes.c_void_p), nmax,
            pool.ctypes.data_as(ctypes.c_void_p), len(pool),
            ctypes.byref(n), ctypes.byref(plen)), "aocs_decode_ao_text")
        vals = [bytes(pool[offs[i]:offs[i] + lens[i]])
                for i in range(n.value)]
        return vals, nulls[:n.value].copy()

    @staticmethod
    def hash_join_spill(build_keys, build_vals, probe_keys,
                        budget_bytes):
        """Spill-tier hash join (unique build keys); returns
        (probe_idx, matched_vals, npartitions)."""
        import numpy as np
        bk = np.ascontiguousarray(build_keys, np.int64)
        bv = np.ascontiguousarray(build_vals, np.int64)
        pk = np.ascontiguousarray(probe_keys, np.int64)
        cap = len(pk) + 1
        oi = np.zeros(cap, np.int64)
        ov = np.zeros(cap, np.int64)
        nm = I64()
        np_ = ctypes.c_int32()
        _check(lib().gg_engine_hash_join_i64_spill(
            bk.ctypes.data_as(ctypes.c_void_p),
            bv.ctypes.data_as(ctypes.c_void_p), len(bk),
            pk.ctypes.data_as(ctypes.c_void_p), len(pk), budget_bytes,
            oi.ctypes.data_as(ctypes.c_void_p),
            ov.ctypes.data_as(ctypes.c_void_p), cap, ctypes.byref(nm),
            ctypes.byref(np_)), "join_spill")
        return oi[:nm.value], ov[:nm.value], np_.value

    @staticmethod
    def hash_groupby_spill(keys, vals, budget_bytes):
        """Spill-tier group-by: partitions staged to host when the
        input exceeds budget_bytes of device memory; returns
        (keys, sums, counts, npartitions)."""
        import numpy as np
        keys = np.ascontiguousarray(keys, np.int64)
        vals = np.ascontiguousarray(vals, np.int64)
        n = len(keys)
        ok = np.zeros(n + 1, np.int64)
        os_ = np.zeros(n + 1, np.int64)
        oc = np.zeros(n + 1, np.int64)
        ng = I64()
        np_ = ctypes.c_int32()
        _check(lib().gg_engine_hash_groupby_i64_spill(
            keys.ctypes.data_as(ctypes.c_void_p),
            vals.ctypes.data_as(ctypes.c_void_p), n, budget_bytes,
            ok.ctypes.data_as(ctypes.c_void_p),
            os_.ctypes.data_as(ctypes.c_void_p),
            oc.ctypes.data_as(ctypes.c_void_p), n + 1,
            ctypes.byref(ng), ctypes.byref(np_)), "groupby_spill")
        g = ng.value
        return ok[:g], os_[:g], oc[:g], np_.value

    @staticmethod
    def text_dict_encode(values, nulls=None, max_dict=4096):
        """GPU dictionary-encode a list of bytes values; returns
        (codes int32, dict list of bytes).  NULLs -> -1."""
        import numpy as np
        n = len(values)
        blob = b"".join(values)
        offs = np.zeros(n, np.uint64)
        lens = np.zeros(n, np.uint32)
        pos = 0
        for i, v in enumerate(values):
            offs[i] = pos
            lens[i] = len(v)
            pos += len(v)
        pool = (np.frombuffer(blob, np.uint8).copy() if blob
                else np.zeros(1, np.uint8))
        nl = (None if nulls is None
              else np.ascontiguousarray(nulls, np.uint8))
        codes = np.zeros(n, np.int32)
        dcap = pos + 16
        dbytes = np.zeros(dcap, np.uint8)
        doffs = np.zeros(max_dict + 1, np.int64)
        nd = ctypes.c_int32()
        _check(lib().gg_engine_text_dict_encode(
            pool.ctypes.data_as(ctypes.c_void_p),
            offs.ctypes.data_as(ctypes.c_void_p),
            lens.ctypes.data_as(ctypes.c_void_p),
            None if nl is None else
            nl.ctypes.data_as(ctypes.c_void_p), n, max_dict,
            codes.ctypes.data_as(ctypes.c_void_p),
            dbytes.ctypes.data_as(ctypes.c_void_p), dcap,
            doffs.ctypes.data_as(ctypes.c_void_p),
            ctypes.byref(nd)), "text_dict_encode")
        d = [bytes(dbytes[doffs[i]:doffs[i + 1]])
             for i in range(nd.value)]
        return codes, d

    @staticmethod
    def motion_chunkify(tuples, max_chunk=8192, append_eos=True):
        """Frame a MemTuple stream as interconnect tuple chunks."""
        import numpy as np
        tuples = np.ascontiguousarray(tuples, np.uint8)
        cap = len(tuples) + (len(tuples) // 16 + 16) * 8 + 64
        out = np.zeros(cap, np.uint8)
        n = I64()
        _check(lib().gg_engine_motion_chunkify(
            tuples.ctypes.data_as(ctypes.c_void_p), len(tuples),
            max_chunk, int(append_eos),
            out.ctypes.data_as(ctypes.c_void_p), cap, ctypes.byref(n)),
            "motion_chunkify")
        return out[:n.value].copy()

    @staticmethod
    def motion_dechunkify(chunks):
        import numpy as np
        chunks = np.ascontiguousarray(chunks, np.uint8)
        out = np.zeros(len(chunks) + 64, np.uint8)
        n, eos = I64(), ctypes.c_int()
        _check(lib().gg_engine_motion_dechunkify(
            chunks.ctypes.data_as(ctypes.c_void_p), len(chunks),
            out.ctypes.data_as(ctypes.c_void_p), len(out),
            ctypes.byref(n), ctypes.byref(eos)), "motion_dechunkify")
        return out[:n.value].copy(), bool(eos.value)

    @staticmethod
    def memtuple_binding_large(attlen, attalign):
        """Host-only restated LARGE binding layout (CPU-testable)."""
        natts = len(attlen)
        al = (ctypes.c_int32 * natts)(*attlen)
        out = (ctypes.c_int32 * (3 + 5 * natts))()
        _check(lib().gg_engine_memtuple_binding_large(
            natts, al, "".join(attalign).encode(), out),
            "memtuple_binding_large")
        meta = (out[0], out[1], out[2])
        per = [tuple(out[3 + i * 5: 8 + i * 5]) for i in range(natts)]
        return meta, per

    @staticmethod
    def memtuple_binding(attlen, attalign):
        """Host-only restated binding layout (CPU-testable)."""
        natts = len(attlen)
        al = (ctypes.c_int32 * natts)(*attlen)
        out = (ctypes.c_int32 * (3 + 5 * natts))()
        _check(lib().gg_engine_memtuple_binding(
            natts, al, "".join(attalign).encode(), out),
            "memtuple_binding")
        meta = (out[0], out[1], out[2])
        per = [tuple(out[3 + i * 5: 8 + i * 5]) for i in range(natts)]
        return meta, per

    @staticmethod
    def memtuple_encode(attlen, attalign, cols, nulls):
        """Bulk GPU encode of column arrays into a MemTuple stream.
        cols[i]: numpy array with itemsize attlen[i], or a list of
        python bytes for attlen -1 (text); nulls[i]: uint8 array or
        None."""
        import numpy as np

        class _TC(ctypes.Structure):
            _fields_ = [("bytes", ctypes.c_void_p),
                        ("offs", ctypes.c_void_p)]

        natts = len(attlen)
        al = (ctypes.c_int32 * natts)(*attlen)
        nrows = len(cols[0])
        carr, keep = [], []
        for l, c in zip(attlen, cols):
            if l == -1:
                blob = b"".join(c)
                offs = np.zeros(nrows + 1, np.int64)
                pos = 0
                for i, v in enumerate(c):
                    pos += len(v)
                    offs[i + 1] = pos
                bts = (np.frombuffer(blob, np.uint8).copy() if blob
                       else np.zeros(1, np.uint8))
                tc = _TC(bts.ctypes.data_as(ctypes.c_void_p).value,
                         offs.ctypes.data_as(ctypes.c_void_p).value)
                keep += [bts, offs, tc]
                carr.append(tc)
            else:
                carr.append(np.ascontiguousarray(c))
        colp = (ctypes.c_void_p * natts)(
            *[ctypes.addressof(c) if isinstance(c, _TC)
              else c.ctypes.data_as(ctypes.c_void_p).value
              for c in carr])
        narr = [None if n is None else
                np.ascontiguousarray(n, np.uint8) for n in nulls]
        nullp = (ctypes.c_void_p * natts)(
            *[0 if n is None else
              n.ctypes.data_as(ctypes.c_void_p).value for n in narr])
        cap = nrows * (8 + sum(8 + max(x, 2) for x in attlen)) + 64
        for l, c in zip(attlen, cols):
            if l == -1:
                cap += sum(len(v) + 8 for v in c)
        out = np.zeros(cap, np.uint8)
        olen = I64()
        _check(lib().gg_engine_memtuple_encode(
            natts, al, "".join(attalign).encode(), colp, nullp, nrows,
            out.ctypes.data_as(ctypes.c_void_p), cap,
            ctypes.byref(olen)), "memtuple_encode")
        return out[:olen.value].copy()

    @staticmethod
    def memtuple_decode(attlen, attalign, stream, cap_rows):
        """Decode a MemTuple stream; text attrs (attlen -1) come back
        as lists of python bytes."""
        import numpy as np

        class _TO(ctypes.Structure):
            _fields_ = [("offs", ctypes.c_void_p),
                        ("lens", ctypes.c_void_p)]

        natts = len(attlen)
        al = (ctypes.c_int32 * natts)(*attlen)
        stream = np.ascontiguousarray(stream, np.uint8)
        dt = {1: np.uint8, 2: np.int16, 4: np.int32, 8: np.int64}
        cols, keep = [], []
        for l in attlen:
            if l == -1:
                offs = np.zeros(cap_rows, np.uint64)
                lens = np.zeros(cap_rows, np.uint32)
                to = _TO(offs.ctypes.data_as(ctypes.c_void_p).value,
                         lens.ctypes.data_as(ctypes.c_void_p).value)
                keep.append((offs, lens))
                cols.append(to)
            else:
                cols.append(np.zeros(cap_rows, dt[l]))
        nulls = [np.zeros(cap_rows, np.uint8) for _ in attlen]
        colp = (ctypes.c_void_p * natts)(
            *[ctypes.addressof(c) if isinstance(c, _TO)
              else c.ctypes.data_as(ctypes.c_void_p).value
              for c in cols])
        nullp = (ctypes.c_void_p * natts)(
            *[n.ctypes.data_as(ctypes.c_void_p).value for n in nulls])
        n = I64()
        _check(lib().gg_engine_memtuple_decode(
            natts, al, "".join(attalign).encode(),
            stream.ctypes.data_as(ctypes.c_void_p), len(stream), colp,
            nullp, cap_rows, ctypes.byref(n)), "memtuple_decode")
        out, ki = [], 0
        for l, c in zip(attlen, cols):
            if l == -1:
                offs, lens = keep[ki]
                ki += 1
                out.append([bytes(stream[int(offs[i]):
                                         int(offs[i]) + int(lens[i])])
                            for i in range(n.value)])
            else:
                out.append(c[:n.value])
        return out, [nl[:n.value] for nl in nulls]

    # ---- general hash group-by (arbitrary int64 keys, SUM+COUNT) ----
    @staticmethod
    def hash_groupby(keys, vals):
        import numpy as np
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        vals = np.ascontiguousarray(vals, dtype=np.int64)
        assert len(keys) == len(vals)
        cap = len(keys) if len(keys) else 1
        ok = np.empty(cap, np.int64)
        os_ = np.empty(cap, np.int64)
        oc = np.empty(cap, np.int64)
        ng = I64()
        _check(lib().gg_engine_hash_groupby_i64(
            keys.ctypes.data_as(ctypes.c_void_p),
            vals.ctypes.data_as(ctypes.c_void_p), len(keys),
            ok.ctypes.data_as(ctypes.c_void_p),
            os_.ctypes.data_as(ctypes.c_void_p),
            oc.ctypes.data_as(ctypes.c_void_p), cap,
            ctypes.byref(ng)), "hash_groupby")
        n = ng.value
        return ok[:n], os_[:n], oc[:n]

    # ---- general sort (ORDER BY operator; LSB radix on GPU) ----
    @staticmethod
    def radix_sort(keys, payload=None, key_bytes=8, descending=False):
        import numpy as np
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        pp = None
        if payload is not None:
            payload = np.ascontiguousarray(payload, dtype=np.uint64)
            assert len(payload) == len(keys)
            pp = payload.ctypes.data_as(ctypes.c_void_p)
        _check(lib().gg_engine_radix_sort_u64(
            keys.ctypes.data_as(ctypes.c_void_p), pp, len(keys),
            key_bytes, 1 if descending else 0), "radix_sort")
        return keys, payload

    # ---- numeric display (product-side finalize) ----
    @staticmethod
    def numeric_str(val128, scale):
        buf = ctypes.create_string_buffer(80)
        lib().gg_engine_numeric_str(val128 & ((1 << 64) - 1), val128 >> 64,
                                    scale, buf)
        return buf.value.decode()

    @staticmethod
    def avg_str(sum128, sum_scale, count):
        buf = ctypes.create_string_buffer(80)
        lib().gg_engine_avg_str(sum128 & ((1 << 64) - 1), sum128 >> 64,
                                sum_scale, count, buf)
        return buf.value.decode()


def _unpack_q1_group(b):
    return {
        "count": int.from_bytes(b[0:8], "little", signed=True),
        "sum_qty_c": int.from_bytes(b[8:16], "little", signed=True),
        "sum_base_c": int.from_bytes(b[16:24], "little", signed=True),
        "sum_dcol_c": int.from_bytes(b[24:32], "little", signed=True),
        "sum_disc4": _i128(int.from_bytes(b[32:40], "little"),
                           int.from_bytes(b[40:48], "little", signed=True)),
        "sum_charge6": _i128(int.from_bytes(b[48:56], "little"),
                             int.from_bytes(b[56:64], "little", signed=True)),
        "returnflag": chr(b[64]),
        "linestatus": chr(b[65]),
    }


def _ascontig(arr, ctype):
    import numpy as np
    want = {"int64": np.int64, "dec64": np.int64, "int32": np.int32,
            "char1": np.uint8}[ctype]
    return np.ascontiguousarray(arr, dtype=want)

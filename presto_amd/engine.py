"""ctypes binding of libpresto_gpu.so (the product C-ABI).

Mirrors include/presto_gpu.h exactly; see that header for the contract and
the reference citations.
"""
import ctypes as C
import pathlib

import numpy as np

_HERE = pathlib.Path(__file__).resolve().parent
_SO = _HERE / "libpresto_gpu.so"

# ---- enums (presto_gpu.h) ----
T_U8, T_I32, T_I64, T_F64, T_VARBIN, T_I128 = 0, 1, 2, 3, 4, 5
(CMP_LT, CMP_LE, CMP_GT, CMP_GE, CMP_EQ, CMP_NE, CMP_CONTAINS,
 CMP_PREFIX, CMP_CONTAINS2, CMP_NOT_CONTAINS2) = range(10)
(PROJ_IDENT, PROJ_DISC_PRICE, PROJ_CHARGE, PROJ_MUL, PROJ_DIV,
 PROJ_KEYSHL, PROJ_SHR, PROJ_SUBDIV, PROJ_KEYSHL_DIV) = range(9)
(AGG_COUNT, AGG_SUM_F64, AGG_SUM_DEC, AGG_SUM_I64,
 AGG_MIN, AGG_MAX) = range(6)
(OP_FILTER_PROJECT, OP_HASH_AGG_SMALL, OP_HASH_BUILD, OP_LOOKUP_JOIN,
 OP_TOPN, OP_PARTITION, OP_GROUPBY_MULTI) = range(1, 8)

_NP_TAG = {np.dtype(np.uint8): T_U8, np.dtype(np.int32): T_I32,
           np.dtype(np.int64): T_I64, np.dtype(np.float64): T_F64}
_TAG_NP = {v: k for k, v in _NP_TAG.items()}


class PgCol(C.Structure):
    _fields_ = [("tag", C.c_int32), ("on_device", C.c_int32),
                ("data", C.c_void_p), ("null_mask", C.c_void_p),
                ("offsets", C.c_void_p), ("dict_ids", C.c_void_p),
                ("dict_n", C.c_int32)]


class PgPage(C.Structure):
    _fields_ = [("n_rows", C.c_int64), ("n_cols", C.c_int32),
                ("cols", PgCol * 32)]


class Pred(C.Structure):
    _fields_ = [("col", C.c_int32), ("op", C.c_int32), ("ival", C.c_int64),
                ("dval", C.c_double), ("sval", C.c_char * 40),
                ("slen", C.c_int32), ("rhs_col", C.c_int32)]


class Proj(C.Structure):
    _fields_ = [("kind", C.c_int32), ("a", C.c_int32), ("b", C.c_int32),
                ("c", C.c_int32)]


class Agg(C.Structure):
    _fields_ = [("func", C.c_int32), ("proj", Proj), ("dec_scale", C.c_int32)]


class PlanFilterProject(C.Structure):
    _fields_ = [("n_preds", C.c_int32), ("preds", Pred * 8),
                ("n_proj", C.c_int32), ("proj", Proj * 16),
                ("semijoin_table", C.c_int64), ("semijoin_col", C.c_int32),
                ("semijoin_anti", C.c_int32)]


class PlanHashAggSmall(C.Structure):
    _fields_ = [("n_preds", C.c_int32), ("preds", Pred * 8),
                ("n_keys", C.c_int32), ("key_col", C.c_int32 * 2),
                ("n_vals", C.c_int32 * 2), ("key_vals", (C.c_uint8 * 8) * 2),
                ("n_aggs", C.c_int32), ("aggs", Agg * 8),
                ("drop_unlisted_keys", C.c_int32)]


class PlanHashBuild(C.Structure):
    _fields_ = [("n_preds", C.c_int32), ("preds", Pred * 8),
                ("key_col", C.c_int32), ("semijoin_table", C.c_int64),
                ("semijoin_col", C.c_int32), ("n_payload", C.c_int32),
                ("payload_col", C.c_int32 * 4), ("capacity_hint", C.c_int64),
                ("key_set_only", C.c_int32), ("dense_array", C.c_int32),
                ("payload_lookup_table", C.c_int64),
                ("payload_lookup_key_col", C.c_int32),
                ("agg_table", C.c_int32), ("pack_bits", C.c_int32),
                ("fill_x10", C.c_int32), ("bitmap_max_key", C.c_int64),
                ("range_group", C.c_int32),
                ("dense_payload_bias", C.c_int32)]


class PlanLookupJoin(C.Structure):
    _fields_ = [("table", C.c_int64), ("n_preds", C.c_int32),
                ("preds", Pred * 8), ("key_col", C.c_int32),
                ("mode", C.c_int32), ("n_emit", C.c_int32),
                ("emit_probe_cols", C.c_int32 * 8), ("proj", Proj),
                ("dec_scale", C.c_int32), ("table2", C.c_int64),
                ("table2_key_col", C.c_int32),
                ("n_group_vals", C.c_int32),
                ("group_vals", C.c_uint8 * 8), ("dec_only", C.c_int32),
                ("dec_min", C.c_int32), ("n_aggs", C.c_int32),
                ("aggs", Agg * 6), ("agg_filter", C.c_int32 * 6),
                ("acc_pack", C.c_int32), ("acc_pack_shift", C.c_int32 * 6),
                ("acc_pack_width", C.c_int32 * 6),
                ("acc_pack_cnt_shift", C.c_int32),
                ("acc_pack_cnt_width", C.c_int32)]


class PlanGroupBy(C.Structure):
    _fields_ = [("n_preds", C.c_int32), ("preds", Pred * 8),
                ("n_keys", C.c_int32), ("key_col", C.c_int32 * 4),
                ("capacity_hint", C.c_int64), ("n_aggs", C.c_int32),
                ("aggs", Agg * 6), ("agg_filter", C.c_int32 * 6)]


class PlanTopN(C.Structure):
    _fields_ = [("limit", C.c_int32), ("val_col", C.c_int32),
                ("date_col", C.c_int32), ("key_col", C.c_int32)]


class PlanPartition(C.Structure):
    _fields_ = [("n_partitions", C.c_int32), ("key_col", C.c_int32),
                ("n_emit", C.c_int32), ("emit_cols", C.c_int32 * 8)]


class _Lib:
    def __init__(self):
        if not _SO.exists():
            raise RuntimeError(
                f"{_SO} not built — run __graft_entry__.build() first "
                "(the product HIP library is required; no CPU fallback)")
        self.c = C.CDLL(str(_SO))
        self.c.pg_last_error.restype = C.c_char_p
        self.c.pg_op_create.argtypes = [C.c_int32, C.c_void_p, C.c_int64,
                                        C.POINTER(C.c_int64)]
        self.c.pg_op_add_input.argtypes = [C.c_int64, C.POINTER(PgPage)]
        self.c.pg_op_get_output.argtypes = [C.c_int64,
                                            C.POINTER(C.POINTER(PgPage))]
        for f in ("pg_op_finish", "pg_op_destroy", "pg_op_needs_input",
                  "pg_op_is_finished"):
            getattr(self.c, f).argtypes = [C.c_int64]
        self.c.pg_op_table.argtypes = [C.c_int64, C.POINTER(C.c_int64)]
        self.c.pg_op_partition_counts.argtypes = [C.c_int64,
                                                  C.POINTER(C.c_int64),
                                                  C.c_int32]
        self.c.pg_table_destroy.argtypes = [C.c_int64]
        self.c.pg_table_reset_acc.argtypes = [C.c_int64]
        self.c.pg_memcpy_d2h.argtypes = [C.c_void_p, C.c_void_p, C.c_int64]
        self.c.pg_memcpy_d2d.argtypes = [C.c_void_p, C.c_void_p, C.c_int64]
        self.c.pg_memcpy_h2d.argtypes = [C.c_void_p, C.c_void_p, C.c_int64]
        self.c.pg_device_sync.argtypes = []

    def err(self):
        return (self.c.pg_last_error() or b"").decode()

    def check(self, st, what):
        if st != 0:
            raise RuntimeError(f"{what}: {self.err()}")


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _Lib()
    return _lib


class Varbin:
    """A host VariableWidthBlock column (VariableWidthBlock.java:48-61):
    concatenated bytes + n+1 int32 offsets."""

    def __init__(self, strings):
        self.offsets = np.zeros(len(strings) + 1, np.int32)
        for i, b in enumerate(strings):
            self.offsets[i + 1] = self.offsets[i] + len(b)
        joined = b"".join(strings)
        self.data = np.frombuffer(joined, np.uint8).copy() if joined \
            else np.empty(0, np.uint8)
        self.n = len(strings)

    def __len__(self):
        return self.n

    def tolist(self):
        return [self.data[self.offsets[i]:self.offsets[i + 1]].tobytes()
                for i in range(self.n)]


class DictVarbin:
    """A host DictionaryBlock over a Varbin dictionary
    (DictionaryBlock.java:60-86): per-position int32 ids into the
    dictionary entries."""

    def __init__(self, dictionary_strings, ids):
        self.dictionary = Varbin(dictionary_strings)
        self.ids = np.ascontiguousarray(ids, np.int32)

    def __len__(self):
        return len(self.ids)


class DeviceVarbin:
    """A VariableWidthBlock resident in HBM: torch-cuda uint8 bytes +
    int32 offsets (+ optional int32 dictionary ids for DictionaryBlock
    columns).  The boundary form of Varbin/DictVarbin once a page has
    been staged."""

    def __init__(self, data, offsets, dict_ids=None, dict_n=0):
        self.data = data          # torch.uint8 cuda tensor
        self.offsets = offsets    # torch.int32 cuda tensor (n+1)
        self.dict_ids = dict_ids  # torch.int32 cuda tensor or None
        self.dict_n = dict_n

    @classmethod
    def from_host(cls, v):
        import torch
        if isinstance(v, DictVarbin):
            return cls(torch.from_numpy(v.dictionary.data).cuda(),
                       torch.from_numpy(v.dictionary.offsets).cuda(),
                       torch.from_numpy(v.ids).cuda(), v.dictionary.n)
        return cls(torch.from_numpy(v.data).cuda(),
                   torch.from_numpy(v.offsets).cuda())

    def __len__(self):
        n = len(self.dict_ids) if self.dict_ids is not None             else len(self.offsets) - 1
        return n


class Page:
    """A Presto Page: named columns backed by numpy (host) or torch-cuda
    (device) arrays (Varbin / DeviceVarbin for variable-width columns).
    Column order is the channel order."""

    def __init__(self, cols, n_rows=None):
        self.names = list(cols.keys())
        self.cols = cols
        first = next(iter(cols.values()))
        self.n_rows = n_rows if n_rows is not None else len(first)

    def channel(self, name):
        return self.names.index(name)

    def to_c(self):
        pg = PgPage()
        pg.n_rows = self.n_rows
        pg.n_cols = len(self.names)
        for i, name in enumerate(self.names):
            a = self.cols[name]
            col = PgCol()
            if isinstance(a, DeviceVarbin):
                col.tag = T_VARBIN
                col.on_device = 1
                col.data = a.data.data_ptr()
                col.offsets = a.offsets.data_ptr()
                if a.dict_ids is not None:
                    col.dict_ids = a.dict_ids.data_ptr()
                    col.dict_n = a.dict_n
            elif isinstance(a, Varbin):
                col.tag = T_VARBIN
                col.on_device = 0
                col.data = a.data.ctypes.data
                col.offsets = a.offsets.ctypes.data
            elif isinstance(a, DictVarbin):
                col.tag = T_VARBIN
                col.on_device = 0
                col.data = a.dictionary.data.ctypes.data
                col.offsets = a.dictionary.offsets.ctypes.data
                col.dict_ids = a.ids.ctypes.data
                col.dict_n = a.dictionary.n
            elif isinstance(a, np.ndarray):
                col.tag = _NP_TAG[a.dtype]
                col.on_device = 0
                col.data = a.ctypes.data
            else:  # torch tensor on cuda
                import torch
                assert isinstance(a, torch.Tensor) and a.is_cuda
                tagmap = {torch.uint8: T_U8, torch.int32: T_I32,
                          torch.int64: T_I64, torch.float64: T_F64}
                col.tag = tagmap[a.dtype]
                col.on_device = 1
                col.data = a.data_ptr()
            col.null_mask = None
            pg.cols[i] = col
        return pg


def _read_output_page(cpage, names=None):
    """Copy an output pg_page (host or device cols) into numpy arrays."""
    L = lib()
    n = cpage.n_rows
    out = {}
    for i in range(cpage.n_cols):
        col = cpage.cols[i]
        if col.tag == T_VARBIN:
            offs = np.empty(n + 1, np.int32)
            if col.on_device:
                L.check(L.c.pg_memcpy_d2h(offs.ctypes.data, col.offsets,
                                          (n + 1) * 4), "d2h")
            else:
                C.memmove(offs.ctypes.data, col.offsets, (n + 1) * 4)
            nb = int(offs[n])
            data = np.empty(nb, np.uint8)
            if nb:
                if col.on_device:
                    L.check(L.c.pg_memcpy_d2h(data.ctypes.data, col.data,
                                              nb), "d2h")
                else:
                    C.memmove(data.ctypes.data, col.data, nb)
            v = Varbin.__new__(Varbin)
            v.offsets, v.data, v.n = offs, data, n
            out[names[i] if names else f"c{i}"] = v
            continue
        dt = _TAG_NP[col.tag]
        a = np.empty(n, dt)
        nbytes = n * dt.itemsize
        if nbytes:
            if col.on_device:
                L.check(L.c.pg_memcpy_d2h(a.ctypes.data, col.data, nbytes),
                        "d2h")
            else:
                C.memmove(a.ctypes.data, col.data, nbytes)
        out[names[i] if names else f"c{i}"] = a
    return out


class Operator:
    """Host mirror of operator/Operator.java:20-102 over the C-ABI."""

    def __init__(self, kind, plan):
        L = lib()
        h = C.c_int64()
        L.check(L.c.pg_op_create(kind, C.byref(plan), C.sizeof(plan),
                                 C.byref(h)), "op_create")
        self.h = h.value
        self.kind = kind

    def needs_input(self):
        return bool(lib().c.pg_op_needs_input(self.h))

    def add_input(self, page: Page):
        L = lib()
        cp = page.to_c()
        L.check(L.c.pg_op_add_input(self.h, C.byref(cp)), "add_input")

    def add_input_raw(self, cpage: PgPage):
        """Feed another operator's raw output page (device pointers pass
        through without copies — pages are borrowed for the call)."""
        L = lib()
        L.check(L.c.pg_op_add_input(self.h, C.byref(cpage)), "add_input")

    def get_output(self, names=None):
        """Returns dict of numpy arrays, or None. (Materializes device
        outputs to host for harness use; device-to-device chaining uses
        get_output_raw.)"""
        L = lib()
        pp = C.POINTER(PgPage)()
        L.check(L.c.pg_op_get_output(self.h, C.byref(pp)), "get_output")
        if not pp:
            return None
        return _read_output_page(pp.contents, names)

    def get_output_raw(self):
        """Returns a snapshot of the raw PgPage descriptor (device pointers
        stay on device) or None.  The column BUFFERS it points at are owned
        by the library: valid until the next get_output on this operator
        (PARTITION: until destroy — its pages alias op-owned buffers)."""
        L = lib()
        pp = C.POINTER(PgPage)()
        L.check(L.c.pg_op_get_output(self.h, C.byref(pp)), "get_output")
        if not pp:
            return None
        snap = PgPage()
        C.memmove(C.byref(snap), pp, C.sizeof(PgPage))
        return snap

    def finish(self):
        lib().check(lib().c.pg_op_finish(self.h), "finish")

    def is_finished(self):
        return bool(lib().c.pg_op_is_finished(self.h))

    def table(self):
        t = C.c_int64()
        lib().check(lib().c.pg_op_table(self.h, C.byref(t)), "table")
        return t.value

    def partition_counts(self, n):
        arr = (C.c_int64 * n)()
        lib().check(lib().c.pg_op_partition_counts(self.h, arr, n),
                    "partition_counts")
        return list(arr)

    def destroy(self):
        h, self.h = self.h, None
        if h is not None:
            lib().c.pg_op_destroy(h)

    def __del__(self):
        # guard on self.h so a stale finalizer can never destroy a live
        # operator if handle reuse is ever introduced
        try:
            if _lib is not None and self.h is not None:
                _lib.c.pg_op_destroy(self.h)
                self.h = None
        except Exception:
            pass

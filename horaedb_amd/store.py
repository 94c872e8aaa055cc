# horaedb_amd/store.py — ctypes binding of the C-ABI (include/horaedb_hx.h).
# Store mirrors the reference's ColumnarStorage trait surface
# (storage.rs:76-89): schema contract fixed by the metric engine, scan via
# scan_agg (hot path) and scan (streaming parity mode).
import ctypes as C
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB = os.path.join(_HERE, "libhoraedb_hx.so")

AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG = 1, 2, 4, 8, 16

_ERRNAMES = {0: "OK", 1: "IO", 2: "FORMAT", 3: "UNSUPPORTED", 4: "NO_GPU",
             5: "HIP", 6: "INVALID", 7: "SCHEMA"}


class HxError(RuntimeError):
    def __init__(self, code, msg):
        super().__init__(f"hx error {_ERRNAMES.get(code, code)}: {msg}")
        self.code = code


class _TimeRange(C.Structure):
    _fields_ = [("start", C.c_int64), ("end", C.c_int64)]


class _SstDesc(C.Structure):
    _fields_ = [("path", C.c_char_p), ("sequence", C.c_uint64)]


class _Pred(C.Structure):
    _fields_ = [("kind", C.c_int32), ("series_ids", C.POINTER(C.c_uint64)),
                ("n_series", C.c_size_t)]


class _ScanSpec(C.Structure):
    _fields_ = [("range", _TimeRange), ("preds", C.POINTER(_Pred)),
                ("n_preds", C.c_size_t), ("ssts", C.POINTER(_SstDesc)),
                ("n_ssts", C.c_size_t), ("projection", C.POINTER(C.c_int32)),
                ("n_projection", C.c_size_t)]


class _AggSpec(C.Structure):
    _fields_ = [("ops", C.c_uint32), ("bucket_ms", C.c_int64)]


class _DeviceSet(C.Structure):
    _fields_ = [("device_ids", C.POINTER(C.c_int32)), ("n_devices", C.c_int32)]


class _ResultTable(C.Structure):
    _fields_ = [("n_groups", C.c_size_t),
                ("series_id", C.POINTER(C.c_uint64)),
                ("bucket", C.POINTER(C.c_int64)),
                ("sum", C.POINTER(C.c_double)),
                ("count", C.POINTER(C.c_uint64)),
                ("vmin", C.POINTER(C.c_double)),
                ("vmax", C.POINTER(C.c_double)),
                ("avg", C.POINTER(C.c_double))]


class _ColBatch(C.Structure):
    _fields_ = [("n_rows", C.c_size_t), ("n_cols", C.c_size_t),
                ("cols", C.POINTER(C.c_void_p)),
                ("col_types", C.POINTER(C.c_int32))]


class _ExecStats(C.Structure):
    _fields_ = [("exec_ms", C.c_double), ("agg_kernel_ms", C.c_double),
                ("decode_kernel_ms", C.c_double), ("rows_scanned", C.c_int64),
                ("rows_matched", C.c_int64), ("bytes_staged", C.c_int64),
                ("stage_ms", C.c_double)]


def _load():
    if not os.path.exists(_LIB):
        raise ImportError(
            f"{_LIB} missing — build it with `make -C horaedb_amd/csrc` "
            "(the HIP product path; no CPU fallback exists)")
    lib = C.CDLL(_LIB)
    lib.hx_last_error.restype = C.c_char_p
    lib.hx_open.argtypes = [C.c_char_p, C.c_int64, C.POINTER(C.c_void_p)]
    lib.hx_close.argtypes = [C.c_void_p]
    lib.hx_find_ssts.argtypes = [C.c_void_p, _TimeRange,
                                 C.POINTER(C.POINTER(_SstDesc)),
                                 C.POINTER(C.c_size_t)]
    lib.hx_prepare.argtypes = [C.c_void_p, C.POINTER(_ScanSpec),
                               C.POINTER(_DeviceSet), C.POINTER(C.c_void_p)]
    lib.hx_prepared_free.argtypes = [C.c_void_p]
    lib.hx_exec_agg.argtypes = [C.c_void_p, C.POINTER(_AggSpec),
                                C.POINTER(C.POINTER(_ResultTable))]
    lib.hx_result_free.argtypes = [C.POINTER(_ResultTable)]
    lib.hx_scan_agg.argtypes = [C.c_void_p, C.POINTER(_ScanSpec),
                                C.POINTER(_AggSpec), C.POINTER(_DeviceSet),
                                C.POINTER(C.POINTER(_ResultTable))]
    lib.hx_get_stats.argtypes = [C.c_void_p, C.POINTER(_ExecStats)]
    lib.hx_scan.argtypes = [C.c_void_p, C.POINTER(_ScanSpec),
                            C.POINTER(_DeviceSet), C.c_void_p, C.c_void_p]
    lib.hx_compact.argtypes = [C.c_void_p, _TimeRange, C.POINTER(_DeviceSet),
                               C.POINTER(C.c_uint64)]
    lib.hx_write.argtypes = [C.c_void_p, C.POINTER(C.c_uint64),
                             C.POINTER(C.c_int64), C.POINTER(C.c_double),
                             C.c_int64, C.c_int32, C.POINTER(C.c_uint64)]
    lib.hx_write_sst.argtypes = [C.c_char_p, C.POINTER(C.c_uint64),
                                 C.POINTER(C.c_int64), C.POINTER(C.c_double),
                                 C.c_uint64, C.c_int64, C.c_int64]
    lib.hx_schema.argtypes = [C.c_void_p, C.c_void_p, C.POINTER(C.c_size_t),
                              C.POINTER(C.c_size_t)]
    lib.hx_catalog_size.argtypes = [C.c_void_p, C.POINTER(C.c_size_t)]
    lib.hx_catalog_entry.argtypes = [C.c_void_p, C.c_size_t,
                                     C.POINTER(C.c_uint64), C.POINTER(C.c_int64),
                                     C.POINTER(C.c_int64), C.POINTER(C.c_int64),
                                     C.POINTER(C.c_int64)]
    return lib


_lib = _load()


def lib_path():
    return _LIB


def _check(code):
    if code != 0:
        raise HxError(code, _lib.hx_last_error().decode())


def _np(ptr, n, dtype):
    if not ptr or n == 0:
        return np.empty(0, dtype)
    return np.ctypeslib.as_array(ptr, shape=(n,)).astype(dtype, copy=True)


class Prepared:
    """A staged scan (column chunks resident in HBM). exec_agg is the timed
    hot path."""

    def __init__(self, store, handle):
        self._store = store
        self._h = handle

    def exec_agg(self, ops=AGG_SUM | AGG_COUNT, bucket_ms=0, copy=True):
        """copy=False: run the full query (result lands in host memory) but
        skip the numpy materialization — for benchmarking loops."""
        spec = _AggSpec(ops=ops, bucket_ms=bucket_ms)
        out = C.POINTER(_ResultTable)()
        _check(_lib.hx_exec_agg(self._h, C.byref(spec), C.byref(out)))
        t = out.contents
        n = t.n_groups
        if not copy:
            _lib.hx_result_free(out)
            return {"n_groups": n}
        res = {"series_id": _np(t.series_id, n, np.uint64)}
        if bucket_ms:
            res["bucket"] = _np(t.bucket, n, np.int64)
        if ops & AGG_SUM:
            res["sum"] = _np(t.sum, n, np.float64)
        if ops & AGG_COUNT:
            res["count"] = _np(t.count, n, np.uint64)
        if ops & AGG_MIN:
            res["vmin"] = _np(t.vmin, n, np.float64)
        if ops & AGG_MAX:
            res["vmax"] = _np(t.vmax, n, np.float64)
        if ops & AGG_AVG:
            res["avg"] = _np(t.avg, n, np.float64)
        _lib.hx_result_free(out)
        return res

    def stats(self):
        st = _ExecStats()
        _check(_lib.hx_get_stats(self._h, C.byref(st)))
        return {f: getattr(st, f) for f, _ in st._fields_}

    def close(self):
        if self._h:
            _lib.hx_prepared_free(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class _TagPred(C.Structure):
    _fields_ = [("tag_key", C.c_char_p), ("tag_value", C.c_char_p)]


class _BytesCol(C.Structure):
    _fields_ = [("offsets", C.POINTER(C.c_int64)),
                ("bytes", C.POINTER(C.c_uint8))]


class Store:
    """MI355X-native ColumnarStorage (scan side). storage.rs:76-89."""

    def __init__(self, store_path, segment_duration_ms=0,
                 update_mode="overwrite"):
        h = C.c_void_p()
        self._update_mode = update_mode
        _check(_lib.hx_open(store_path.encode(), segment_duration_ms,
                            C.byref(h)))
        self._h = h
        if update_mode == "append":
            _check(_lib.hx_set_update_mode(self._h, 1))
        elif update_mode != "overwrite":
            raise ValueError("update_mode: 'overwrite' or 'append'")

    def close(self):
        if self._h:
            _lib.hx_close(self._h)
            self._h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def schema(self):
        """ColumnarStorage::schema (storage.rs:76-89): column layout with
        PK/builtin markers (types.rs:150-240 contract)."""

        class _Col(C.Structure):
            _fields_ = [("name", C.c_char_p), ("col_type", C.c_int32),
                        ("is_primary_key", C.c_int32),
                        ("is_builtin", C.c_int32)]

        out = C.POINTER(_Col)()
        n = C.c_size_t()
        npk = C.c_size_t()
        _check(_lib.hx_schema(self._h, C.cast(C.byref(out), C.c_void_p),
                              C.byref(n), C.byref(npk)))
        tnames = {0: "u64", 1: "i64", 2: "f64"}
        cols = [{"name": out[i].name.decode(),
                 "type": tnames[out[i].col_type],
                 "primary_key": bool(out[i].is_primary_key),
                 "builtin": bool(out[i].is_builtin)} for i in range(n.value)]
        return {"columns": cols, "num_primary_keys": npk.value}

    def catalog(self):
        n = C.c_size_t()
        _check(_lib.hx_catalog_size(self._h, C.byref(n)))
        out = []
        for i in range(n.value):
            seq = C.c_uint64()
            rows = C.c_int64()
            tmin = C.c_int64()
            tmax = C.c_int64()
            nrg = C.c_int64()
            _check(_lib.hx_catalog_entry(self._h, i, C.byref(seq), C.byref(rows),
                                         C.byref(tmin), C.byref(tmax),
                                         C.byref(nrg)))
            out.append({"seq": seq.value, "n_rows": rows.value,
                        "ts_min": tmin.value, "ts_max": tmax.value,
                        "n_row_groups": nrg.value})
        return out

    def find_ssts(self, ts_range):
        """Manifest::find_ssts (manifest/mod.rs:165-172)."""
        out = C.POINTER(_SstDesc)()
        n = C.c_size_t()
        _check(_lib.hx_find_ssts(self._h, _TimeRange(*ts_range), C.byref(out),
                                 C.byref(n)))
        return [(out[i].path.decode(), out[i].sequence) for i in range(n.value)]

    def _spec(self, ts_range, series_in=None):
        # keepalive is CALL-LOCAL (returned, held by the caller until the C
        # call returns): a Store serves prepares from many threads
        spec = _ScanSpec()
        spec.range = _TimeRange(*ts_range)
        keep = []
        if series_in is not None:
            arr = np.ascontiguousarray(np.asarray(sorted(set(int(s) for s in series_in)),
                                                  dtype=np.uint64))
            pred = _Pred(kind=1,
                         series_ids=arr.ctypes.data_as(C.POINTER(C.c_uint64)),
                         n_series=len(arr))
            preds = (_Pred * 1)(pred)
            spec.preds = preds
            spec.n_preds = 1
            keep += [arr, preds]
        return spec, keep

    def _devset(self, devices, keep):
        if devices is None:
            return None
        ids = (C.c_int32 * len(devices))(*devices)
        ds = _DeviceSet(device_ids=ids, n_devices=len(devices))
        keep += [ids, ds]
        return ds

    def prepare(self, ts_range, series_in=None, devices=None,
                sst_subset=None):
        """sst_subset: list of (path, seq) to scan (hx_scan_spec.ssts — the
        per-GPU shard of BASELINE configs 4/5); None = all overlapping."""
        spec, keep = self._spec(ts_range, series_in)
        if sst_subset is not None:
            arr = (_SstDesc * len(sst_subset))()
            enc = [p.encode() for p, _ in sst_subset]
            for i, ((p, q), pe) in enumerate(zip(sst_subset, enc)):
                arr[i].path = pe
                arr[i].sequence = q
            spec.ssts = arr
            spec.n_ssts = len(sst_subset)
            keep += [arr, enc]
        ds = self._devset(devices, keep)
        out = C.c_void_p()
        _check(_lib.hx_prepare(self._h, C.byref(spec),
                               C.byref(ds) if ds else None, C.byref(out)))
        del keep
        return Prepared(self, out)

    def scan_agg(self, ts_range, ops=AGG_SUM | AGG_COUNT, bucket_ms=0,
                 series_in=None, devices=None):
        """One-shot scan+aggregate (prepare + exec + release)."""
        with self.prepare(ts_range, series_in, devices) as p:
            return p.exec_agg(ops=ops, bucket_ms=bucket_ms)

    def index_write(self, tag_keys, tag_values, tsids, metric_id=None):
        """Append one inverted-index SST (rfc:86-137 index table): rows
        (metric_id, tag_key, tag_value, tsid), writer-sorted by
        (tag_key, tag_value, tsid). Returns the index file sequence."""
        n = len(tsids)
        assert len(tag_keys) == n and len(tag_values) == n
        tk = (C.c_char_p * n)(*[k.encode() if isinstance(k, str) else k
                                for k in tag_keys])
        tv = (C.c_char_p * n)(*[v.encode() if isinstance(v, str) else v
                                for v in tag_values])
        tsids = np.ascontiguousarray(tsids, dtype=np.uint64)
        mid = None
        midp = None
        if metric_id is not None:
            mid = np.ascontiguousarray(metric_id, dtype=np.uint64)
            midp = mid.ctypes.data_as(C.POINTER(C.c_uint64))
        seq = C.c_uint64()
        _check(_lib.hx_index_write(
            self._h, midp, tk, tv,
            tsids.ctypes.data_as(C.POINTER(C.c_uint64)), n, C.byref(seq)))
        return int(seq.value)

    def index_query(self, preds, combine="and", device=0):
        """Tag-equality predicates -> sorted distinct TSID set (GPU
        postings filter + set combine). preds: [(key, value), ...]."""
        n = len(preds)
        arr = (_TagPred * n)()
        keep = []
        for i, (k, v) in enumerate(preds):
            kb = k.encode() if isinstance(k, str) else k
            vb = v.encode() if isinstance(v, str) else v
            keep += [kb, vb]
            arr[i].tag_key = kb
            arr[i].tag_value = vb
        out = C.POINTER(C.c_uint64)()
        n_out = C.c_size_t()
        _check(_lib.hx_index_query(self._h, arr, n,
                                   1 if combine == "and" else 0, device,
                                   C.byref(out), C.byref(n_out)))
        try:
            res = np.array(_np(out, n_out.value, np.uint64), copy=True) \
                if n_out.value else np.empty(0, np.uint64)
        finally:
            _lib.hx_tsids_free(out)
        return res

    def write(self, series, ts, value, enable_check=True):
        """ColumnarStorage::write (storage.rs:76-89): stable PK sort + new
        SST + catalog add. Returns the new file's sequence."""
        series = np.ascontiguousarray(series, dtype=np.uint64)
        ts_a = np.ascontiguousarray(ts, dtype=np.int64)
        value = np.ascontiguousarray(value, dtype=np.float64)
        out = C.c_uint64()
        _check(_lib.hx_write(
            self._h, series.ctypes.data_as(C.POINTER(C.c_uint64)),
            ts_a.ctypes.data_as(C.POINTER(C.c_int64)),
            value.ctypes.data_as(C.POINTER(C.c_double)), len(series),
            1 if enable_check else 0, C.byref(out)))
        return out.value

    def compact_files(self, input_seqs, devices=None):
        """General compaction of an EXPLICIT input set (executor
        Task{inputs}, keep_builtin): dedups among the named files only;
        the output SST retains per-row __seq__ values."""
        arr = (C.c_uint64 * len(input_seqs))(*input_seqs)
        keep = []
        ds = self._devset(devices, keep)
        seq = C.c_uint64()
        _check(_lib.hx_compact_files(self._h, arr, len(input_seqs),
                                     C.byref(ds) if ds else None,
                                     C.byref(seq)))
        del keep
        return int(seq.value)

    def compact(self, ts_range, devices=None):
        """ColumnarStorage::compact (storage.rs:76-89): GPU-merge the
        ts-overlap closure of the range's SSTs into one new SST. Returns the
        new file's sequence (0 = nothing to compact)."""
        ds = self._devset(devices, [])
        out = C.c_uint64()
        _check(_lib.hx_compact(self._h, _TimeRange(*ts_range),
                               C.byref(ds) if ds else None, C.byref(out)))
        return out.value

    def scan(self, ts_range, series_in=None, projection=None, devices=None,
             batch_limit=None):
        """Streaming parity mode (hx_scan): the merged, deduplicated row
        stream itself — ColumnarStorage::scan semantics (storage.rs:335-370).
        Returns dict of concatenated column arrays in stream order.
        batch_limit: stop the stream (callback returns nonzero — the
        header's early-stop contract) after that many batches."""
        spec, keep = self._spec(ts_range, series_in)
        if projection is not None:
            parr = (C.c_int32 * len(projection))(*projection)
            spec.projection = parr
            spec.n_projection = len(projection)
            keep.append(parr)
        ds = self._devset(devices, keep)
        chunks = []

        _BATCH = C.CFUNCTYPE(C.c_int32, C.c_void_p, C.POINTER(_ColBatch))

        def on_batch(ctx, bp):
            b = bp.contents
            cols = []
            for i in range(b.n_cols):
                t = b.col_types[i]
                if t == 3:  # Binary value column (hx_bytes_col)
                    bcp = C.cast(b.cols[i], C.POINTER(_BytesCol)).contents
                    offs = np.ctypeslib.as_array(
                        bcp.offsets, shape=(b.n_rows + 1,)).copy()
                    total = int(offs[-1])
                    data = (np.ctypeslib.as_array(
                        bcp.bytes, shape=(total,)).copy()
                        if total else np.empty(0, np.uint8))
                    vals = np.empty(b.n_rows, dtype=object)
                    db = data.tobytes()
                    for r in range(b.n_rows):
                        vals[r] = db[offs[r]:offs[r + 1]]
                    cols.append(vals)
                    continue
                dt = {0: np.uint64, 1: np.int64, 2: np.float64}[t]
                ptr = C.cast(b.cols[i], C.POINTER(C.c_uint64))
                arr = np.ctypeslib.as_array(ptr, shape=(b.n_rows,)).copy()
                cols.append(arr.view(dt))
            chunks.append(cols)
            if batch_limit is not None and len(chunks) >= batch_limit:
                return 1
            return 0

        def on_batch_safe(ctx, bp):
            # exceptions must not cross the C callback boundary
            try:
                return on_batch(ctx, bp)
            except Exception:
                return 1  # stop the stream

        cb = _BATCH(on_batch_safe)
        _check(_lib.hx_scan(self._h, C.byref(spec),
                            C.byref(ds) if ds else None,
                            C.cast(cb, C.c_void_p), None))
        ncols = len(projection) if projection is not None else 3
        names = ["series_id", "timestamp", "value"]
        sel = projection if projection is not None else [0, 1, 2]
        out = {}
        for i in range(ncols):
            parts = [c[i] for c in chunks]
            out[names[sel[i]]] = (np.concatenate(parts) if parts else
                                  np.empty(0))
        return out


def write_sst_native(path, series, ts, value, seq, row_group=8192):
    """Native C++ SST writer (the compaction output path), exposed for
    tests: writes the reference metric-SST layout without pyarrow."""
    series = np.ascontiguousarray(series, dtype=np.uint64)
    ts = np.ascontiguousarray(ts, dtype=np.int64)
    value = np.ascontiguousarray(value, dtype=np.float64)
    n = len(series)
    _check(_lib.hx_write_sst(
        path.encode(), series.ctypes.data_as(C.POINTER(C.c_uint64)),
        ts.ctypes.data_as(C.POINTER(C.c_int64)),
        value.ctypes.data_as(C.POINTER(C.c_double)), seq, n, row_group))

"""ctypes bindings for libvega_gpu.so (the C-ABI in include/vega_gpu.h).

Two layers, mirroring the ABI:
  - VegaContext: the RDD-handle API (single process, one GPU) — the drop-in
    surface a Rust host would bind.
  - dev_*: device-pointer entries for the rank-per-GPU path; they take torch
    CUDA tensors (device memory + stream plumbing only — all compute is in
    the hand-written HIP kernels).

This module FAILS LOUDLY if the HIP library is missing — no CPU fallback.
"""
import ctypes
import os

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
LIBPATH = os.path.join(HERE, "csrc", "libvega_gpu.so")
LIBPATH_TORCH = os.path.join(HERE, "csrc", "libvega_gpu_torch.so")

OP_SUM_I64 = 0
OP_COUNT = 1
OP_SUM_F64 = 2
OP_MIN_I64 = 3
OP_MAX_I64 = 4

MAP_VALUES_ADD = 0
MAP_VALUES_MUL = 1
MAP_KEYS_ADD = 2
MAP_SWAP = 3
PRED_KEY_MOD_EQ = 0
PRED_VAL_GT = 1
PRED_KEY_IN_RANGE = 2

_lib = None


class VegaGpuError(RuntimeError):
    pass


def lib():
    global _lib
    if _lib is None:
        # Prefer the torch-runtime-linked variant and load torch FIRST, so
        # exactly one HIP runtime (torch's bundled one) lives in the process;
        # torch tensors/streams and our kernels then share it.
        path = LIBPATH
        if os.path.exists(LIBPATH_TORCH):
            try:
                import torch  # noqa: F401  (loads its libamdhip64.so)
                path = LIBPATH_TORCH
            except ImportError:
                pass
        if not os.path.exists(path):
            raise VegaGpuError(
                f"{path} not built — run __graft_entry__.build() (hipcc "
                "--offload-arch=gfx950); the GPU path has no fallback")
        _lib = ctypes.CDLL(path)
        _lib.vega_dev_ws_bytes.restype = ctypes.c_size_t
        _lib.vega_dev_ws_bytes.argtypes = [ctypes.c_uint64]
        _lib.vega_gpu_last_error.restype = ctypes.c_char_p
    return _lib


def _np(a, dtype=np.int64):
    return np.ascontiguousarray(a, dtype=dtype)


def _pp(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def _check(rc, what, ctx=None):
    if rc != 0:
        extra = b""
        if ctx is not None:
            extra = lib().vega_gpu_last_error(ctx)
        raise VegaGpuError(f"{what} failed rc={rc} {extra!r}")


class VegaContext:
    """Mirrors Context (context.rs:147-164) + PairRdd ops for one GPU."""

    def __init__(self, ngpus=1):
        self._c = ctypes.c_void_p()
        _check(lib().vega_gpu_init(ngpus, ctypes.byref(self._c)), "init")

    def close(self):
        if self._c:
            lib().vega_gpu_shutdown(self._c)
            self._c = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    # --- construction (context.rs:406-442 make_rdd/parallelize) ---
    def make_rdd(self, keys, vals, nparts=4):
        h = ctypes.c_uint64()
        if np.asarray(vals).dtype == np.float64:
            k = _np(keys)
            v = _np(vals, np.float64)
            _check(lib().vega_gpu_make_rdd_f64(self._c, _pp(k), _pp(v),
                                               ctypes.c_uint64(len(k)),
                                               ctypes.c_uint32(nparts),
                                               ctypes.byref(h)), "make_rdd_f64", self._c)
        else:
            k = _np(keys)
            v = _np(vals)
            _check(lib().vega_gpu_make_rdd(self._c, _pp(k), _pp(v),
                                           ctypes.c_uint64(len(k)),
                                           ctypes.c_uint32(nparts),
                                           ctypes.byref(h)), "make_rdd", self._c)
        return Rdd(self, h.value, np.asarray(vals).dtype)

    def gen_rdd_uniform(self, n, seed, key_bits=63, start=0, nparts=256):
        h = ctypes.c_uint64()
        _check(lib().vega_gpu_gen_rdd_uniform(
            self._c, ctypes.c_uint64(n), ctypes.c_uint64(seed),
            ctypes.c_int(key_bits), ctypes.c_uint64(start),
            ctypes.c_uint32(nparts), ctypes.byref(h)), "gen_rdd", self._c)
        return Rdd(self, h.value, np.dtype(np.int64))

    def synchronize(self):
        _check(lib().vega_gpu_synchronize(self._c), "sync", self._c)

    def set_profiling(self, on):
        lib().vega_gpu_set_profiling(self._c, 1 if on else 0)

    def kernel_stats(self):
        import json
        buf = ctypes.create_string_buffer(1 << 16)
        _check(lib().vega_gpu_kernel_stats(self._c, buf, len(buf)), "stats")
        return json.loads(buf.value.decode())


class Rdd:
    def __init__(self, ctx, handle, vdtype):
        self.ctx = ctx
        self.h = handle
        self.vdtype = np.dtype(vdtype)

    # --- PairRdd ops (pair_rdd.rs names) ---
    def reduce_by_key(self, op=OP_SUM_I64, nparts=256):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_reduce_by_key(self.ctx._c, ctypes.c_uint64(self.h),
                                            ctypes.c_int(op), ctypes.c_uint32(nparts),
                                            ctypes.byref(out)),
               "reduce_by_key", self.ctx._c)
        return Rdd(self.ctx, out.value,
                   np.float64 if op == OP_SUM_F64 else np.int64)

    def map(self, op, p0=0):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_map(self.ctx._c, ctypes.c_uint64(self.h),
                                  ctypes.c_int(op), ctypes.c_int64(p0),
                                  ctypes.byref(out)), "map", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def filter(self, pred, p0=0, p1=0):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_filter(self.ctx._c, ctypes.c_uint64(self.h),
                                     ctypes.c_int(pred), ctypes.c_int64(p0),
                                     ctypes.c_int64(p1), ctypes.byref(out)),
               "filter", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def distinct(self, nparts=256):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_distinct(self.ctx._c, ctypes.c_uint64(self.h),
                                       ctypes.c_uint32(nparts), ctypes.byref(out)),
               "distinct", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def count_by_value(self, nparts=256):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_count_by_value(self.ctx._c, ctypes.c_uint64(self.h),
                                             ctypes.c_uint32(nparts), ctypes.byref(out)),
               "count_by_value", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def group_by_key(self):
        """Full groups (pair_rdd.rs:35-52, aggregator.rs:33-53) materialized
        IN THE ENGINE (vega_gpu_group_by_key): returns (keys, offsets,
        values) — values in grouped order, value order within a group = row
        order (stable grouping sort)."""
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_group_by_key(self.ctx._c, ctypes.c_uint64(self.h),
                                           ctypes.c_uint32(256), ctypes.byref(out)),
               "group_by_key", self.ctx._c)
        g = Rdd(self.ctx, out.value, self.vdtype)
        nk = ctypes.c_uint64(0)
        nvals = ctypes.c_uint64(0)
        _check(lib().vega_gpu_collect_groups(self.ctx._c, ctypes.c_uint64(g.h),
                                             None, None, None,
                                             ctypes.byref(nk), ctypes.byref(nvals)),
               "collect_groups size", self.ctx._c)
        keys = np.empty(nk.value, dtype=np.int64)
        offsets = np.empty(nk.value + 1, dtype=np.uint64)
        values = np.empty(nvals.value, dtype=self.vdtype)
        _check(lib().vega_gpu_collect_groups(self.ctx._c, ctypes.c_uint64(g.h),
                                             _pp(keys), _pp(offsets), _pp(values),
                                             ctypes.byref(nk), ctypes.byref(nvals)),
               "collect_groups", self.ctx._c)
        g.free()
        return keys, offsets.astype(np.int64), values

    def cogroup(self, other):
        """cogroup (pair_rdd.rs:123-155): for every key in either side the
        (Vec<V>, Vec<W>) ranges. Returns (keys, offa, lena, offb, lenb,
        vala, valb)."""
        na, nb = self.count(), other.count()
        cap = na + nb
        keys = np.empty(max(cap, 1), dtype=np.int64)
        offa = np.empty(max(cap, 1), dtype=np.uint64)
        lena = np.empty(max(cap, 1), dtype=np.uint64)
        offb = np.empty(max(cap, 1), dtype=np.uint64)
        lenb = np.empty(max(cap, 1), dtype=np.uint64)
        vala = np.empty(max(na, 1), dtype=np.int64)
        valb = np.empty(max(nb, 1), dtype=np.int64)
        nk = ctypes.c_uint64(0)
        _check(lib().vega_gpu_cogroup_collect(
            self.ctx._c, ctypes.c_uint64(self.h), ctypes.c_uint64(other.h),
            _pp(keys), _pp(offa), _pp(lena), _pp(offb), _pp(lenb),
            _pp(vala), _pp(valb), ctypes.c_uint64(cap), ctypes.byref(nk)),
            "cogroup", self.ctx._c)
        n = nk.value
        return (keys[:n], offa[:n].astype(np.int64), lena[:n].astype(np.int64),
                offb[:n].astype(np.int64), lenb[:n].astype(np.int64),
                vala[:na], valb[:nb])

    def intersection(self, other, nparts=256):
        """distinct keys present in both sides (rdd.rs set semantics)"""
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_intersection(self.ctx._c, ctypes.c_uint64(self.h),
                                           ctypes.c_uint64(other.h),
                                           ctypes.c_uint32(nparts), ctypes.byref(out)),
               "intersection", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def subtract(self, other, nparts=256):
        """distinct keys of self absent from other"""
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_subtract(self.ctx._c, ctypes.c_uint64(self.h),
                                       ctypes.c_uint64(other.h),
                                       ctypes.c_uint32(nparts), ctypes.byref(out)),
               "subtract", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def group_count(self, nparts=256):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_group_count(self.ctx._c, ctypes.c_uint64(self.h),
                                          ctypes.c_uint32(nparts), ctypes.byref(out)),
               "group_count", self.ctx._c)
        return Rdd(self.ctx, out.value, np.int64)

    def sort_by_key(self):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_sort_by_key(self.ctx._c, ctypes.c_uint64(self.h),
                                          ctypes.byref(out)), "sort_by_key", self.ctx._c)
        return Rdd(self.ctx, out.value, self.vdtype)

    def join(self, other, nparts=256):
        out = ctypes.c_uint64()
        _check(lib().vega_gpu_join(self.ctx._c, ctypes.c_uint64(self.h),
                                   ctypes.c_uint64(other.h), ctypes.c_uint32(nparts),
                                   ctypes.byref(out)), "join", self.ctx._c)
        r = Rdd(self.ctx, out.value, np.int64)
        r.is_join = True
        return r

    def collect_join(self):
        n = ctypes.c_uint64(self.count())
        k = np.empty(n.value, dtype=np.int64)
        va = np.empty(n.value, dtype=np.int64)
        vb = np.empty(n.value, dtype=np.int64)
        _check(lib().vega_gpu_collect_join(self.ctx._c, ctypes.c_uint64(self.h),
                                           _pp(k), _pp(va), _pp(vb), ctypes.byref(n)),
               "collect_join", self.ctx._c)
        return k[:n.value], va[:n.value], vb[:n.value]

    # --- actions (rdd.rs collect/count) ---
    def count(self):
        n = ctypes.c_uint64()
        _check(lib().vega_gpu_count(self.ctx._c, ctypes.c_uint64(self.h),
                                    ctypes.byref(n)), "count", self.ctx._c)
        return n.value

    def collect(self):
        n = ctypes.c_uint64(self.count())
        k = np.empty(n.value, dtype=np.int64)
        v = np.empty(n.value, dtype=self.vdtype)
        _check(lib().vega_gpu_collect(self.ctx._c, ctypes.c_uint64(self.h),
                                      _pp(k), _pp(v), ctypes.byref(n)),
               "collect", self.ctx._c)
        return k[:n.value], v[:n.value]

    def free(self):
        lib().vega_gpu_free_rdd(self.ctx._c, ctypes.c_uint64(self.h))


# ---------------- device-pointer API over torch tensors ----------------

def _t(t):
    return ctypes.c_void_p(t.data_ptr())


def _stream():
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def ws_bytes(n):
    return int(lib().vega_dev_ws_bytes(ctypes.c_uint64(n)))


def alloc_ws(n, device="cuda"):
    import torch
    return torch.empty(ws_bytes(n), dtype=torch.uint8, device=device)


def dev_gen_uniform(keys_t, vals_t, seed, key_bits=63, start=0):
    n = keys_t.numel()
    _check(lib().vega_dev_gen_uniform_i64(_stream(), _t(keys_t), _t(vals_t),
                                          ctypes.c_uint64(n), ctypes.c_uint64(seed),
                                          ctypes.c_int(key_bits), ctypes.c_uint64(start)),
           "dev_gen")


def dev_gen_uniform_f64(keys_t, vals_t, seed, key_bits=63, start=0):
    n = keys_t.numel()
    _check(lib().vega_dev_gen_uniform_f64(_stream(), _t(keys_t), _t(vals_t),
                                          ctypes.c_uint64(n), ctypes.c_uint64(seed),
                                          ctypes.c_int(key_bits), ctypes.c_uint64(start)),
           "dev_gen_f64")


def dev_partition(keys_t, vals_t, nparts, out_k, out_v, ws_t):
    n = keys_t.numel()
    counts = np.zeros(nparts, dtype=np.uint64)
    _check(lib().vega_dev_partition_i64(_stream(), _t(keys_t), _t(vals_t),
                                        ctypes.c_uint64(n), ctypes.c_uint32(nparts),
                                        _t(out_k), _t(out_v), _pp(counts),
                                        _t(ws_t), ctypes.c_size_t(ws_t.numel())),
           "dev_partition")
    return counts


def dev_partition_range(keys_t, vals_t, splitters_t, out_k, out_v, ws_t):
    """range partition: bucket = #splitters <= key; splitters_t is a device
    tensor of nparts-1 ascending i64 splitters"""
    n = keys_t.numel()
    nparts = splitters_t.numel() + 1
    counts = np.zeros(nparts, dtype=np.uint64)
    _check(lib().vega_dev_partition_range_i64(
        _stream(), _t(keys_t), _t(vals_t), ctypes.c_uint64(n),
        ctypes.c_uint32(nparts), _t(splitters_t), _t(out_k), _t(out_v),
        _pp(counts), _t(ws_t), ctypes.c_size_t(ws_t.numel())),
        "dev_partition_range")
    return counts


def dev_sort_reduce(keys_t, vals_t, op, out_k, out_v, ws_t):
    n = keys_t.numel()
    nout = ctypes.c_uint64()
    _check(lib().vega_dev_sort_reduce(_stream(), _t(keys_t), _t(vals_t),
                                      ctypes.c_uint64(n), ctypes.c_int(op),
                                      _t(out_k), _t(out_v), ctypes.byref(nout),
                                      _t(ws_t), ctypes.c_size_t(ws_t.numel())),
           "dev_sort_reduce")
    return nout.value


def dev_sort_pairs(keys_t, vals_t, ws_t):
    n = keys_t.numel()
    _check(lib().vega_dev_sort_pairs_i64(_stream(), _t(keys_t), _t(vals_t),
                                         ctypes.c_uint64(n), _t(ws_t),
                                         ctypes.c_size_t(ws_t.numel())),
           "dev_sort_pairs")


def dev_group_pairs(keys_t, vals_t, ws_t):
    """in-place grouping-order sort; returns the order tag (4 = (h32,key)
    lex usable with dev_join_grouped mode 2; 0 = full unsigned-key order)"""
    tag = ctypes.c_int(0)
    _check(lib().vega_dev_group_pairs_i64(
        _stream(), _t(keys_t), _t(vals_t), ctypes.c_uint64(keys_t.numel()),
        ctypes.byref(tag), _t(ws_t), ctypes.c_size_t(ws_t.numel())), "dev_group_pairs")
    return tag.value


def dev_join_grouped(ak, av, bk, bv, order_mode, out_k, out_va, out_vb, ws_t):
    nout = ctypes.c_uint64()
    _check(lib().vega_dev_join_grouped(
        _stream(), _t(ak), _t(av), ctypes.c_uint64(ak.numel()),
        _t(bk), _t(bv), ctypes.c_uint64(bk.numel()), ctypes.c_int(order_mode),
        _t(out_k), _t(out_va), _t(out_vb), ctypes.c_uint64(out_k.numel()),
        ctypes.byref(nout), _t(ws_t), ctypes.c_size_t(ws_t.numel())),
        "dev_join_grouped")
    return nout.value


def dev_join_sorted(ak, av, bk, bv, out_k, out_va, out_vb, ws_t):
    """sort-merge inner join of two KEY-SORTED sides; returns rows emitted"""
    nout = ctypes.c_uint64()
    _check(lib().vega_dev_join_sorted(
        _stream(), _t(ak), _t(av), ctypes.c_uint64(ak.numel()),
        _t(bk), _t(bv), ctypes.c_uint64(bk.numel()),
        _t(out_k), _t(out_va), _t(out_vb), ctypes.c_uint64(out_k.numel()),
        ctypes.byref(nout), _t(ws_t), ctypes.c_size_t(ws_t.numel())),
        "dev_join_sorted")
    return nout.value


def dev_checksum(keys_t, vals_t, ws_t):
    n = keys_t.numel()
    s = ctypes.c_uint64()
    _check(lib().vega_dev_checksum_pairs(_stream(), _t(keys_t), _t(vals_t),
                                         ctypes.c_uint64(n), ctypes.byref(s),
                                         _t(ws_t), ctypes.c_size_t(ws_t.numel())),
           "dev_checksum")
    return s.value


def prof_enable(on=True):
    lib().vega_prof_enable(1 if on else 0)


def prof_stats():
    import json
    buf = ctypes.create_string_buffer(1 << 16)
    _check(lib().vega_prof_stats(buf, len(buf)), "prof_stats")
    return json.loads(buf.value.decode())

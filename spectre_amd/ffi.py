"""ctypes binding over libspectre_gpu.so (the product C ABI).

Mirrors include/spectre_gpu.h one-to-one. Compute entry points require a
visible AMD GPU; only `combine_partials` (host-side final reduction) and
`version` work without one — exactly the contract of the C library.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
# SPECTRE_GPU_LIB overrides the library path (tuning-variant A/B builds)
_LIB = os.environ.get("SPECTRE_GPU_LIB",
                      os.path.join(_HERE, "libspectre_gpu.so"))

SCALARS_MONTGOMERY = 0
SCALARS_CANONICAL = 1
WINDOW_BITS = 16  # = SPECTRE_MSM_WINDOW_BITS
NUM_WINDOWS = 16  # = SPECTRE_MSM_NUM_WINDOWS
NUM_BUCKETS = NUM_WINDOWS * (1 << (WINDOW_BITS - 1))
PARTIALS_BYTES = NUM_WINDOWS * 96  # per-shard Jacobian window sums

_lib = None


def lib_path() -> str:
    return _LIB


def load_library() -> ctypes.CDLL:
    """dlopen the product library and declare signatures. Raises if missing —
    build with `make -C spectre_amd/csrc` (or __graft_entry__.build())."""
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB):
        raise RuntimeError(
            f"libspectre_gpu.so not found at {_LIB}; run __graft_entry__.build()"
        )
    lib = ctypes.CDLL(_LIB)
    c = ctypes
    lib.spectre_gpu_init.restype = c.c_void_p
    lib.spectre_gpu_init.argtypes = [c.c_int, c.POINTER(c.c_int)]
    lib.spectre_gpu_destroy.argtypes = [c.c_void_p]
    lib.spectre_gpu_last_error.restype = c.c_char_p
    lib.spectre_gpu_version.restype = c.c_char_p
    lib.spectre_gpu_device_count.restype = c.c_int
    lib.spectre_gpu_device_count.argtypes = [c.c_void_p]
    lib.spectre_gpu_msm_g1.restype = c.c_int
    lib.spectre_gpu_msm_g1.argtypes = [
        c.c_void_p, c.c_uint64, c.c_void_p, c.c_void_p, c.c_uint64,
        c.c_uint32, c.c_int, c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_batch.restype = c.c_int
    lib.spectre_gpu_msm_g1_batch.argtypes = [
        c.c_void_p, c.c_uint64, c.c_void_p, c.c_void_p, c.c_uint32,
        c.c_uint64, c.c_uint32, c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_batch_device.restype = c.c_int
    lib.spectre_gpu_msm_g1_batch_device.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint32, c.c_uint64,
        c.c_uint32, c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_device.restype = c.c_int
    lib.spectre_gpu_msm_g1_device.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_shard_device.restype = c.c_int
    lib.spectre_gpu_msm_g1_shard_device.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_shard_device_timed.restype = c.c_int
    lib.spectre_gpu_msm_g1_shard_device_timed.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_void_p, c.POINTER(c.c_double),
    ]
    lib.spectre_gpu_msm_g1_shard_device_async.restype = c.c_int
    lib.spectre_gpu_msm_g1_shard_device_async.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_void_p, c.POINTER(c.c_int),
    ]
    lib.spectre_gpu_msm_slot_wait.restype = c.c_int
    lib.spectre_gpu_msm_slot_wait.argtypes = [c.c_void_p, c.c_int, c.c_int]
    lib.spectre_gpu_msm_g1_shard_windows_device.restype = c.c_int
    lib.spectre_gpu_msm_g1_shard_windows_device.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_uint32, c.c_uint32, c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_shard_windows_device_async.restype = c.c_int
    lib.spectre_gpu_msm_g1_shard_windows_device_async.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_void_p, c.c_uint64, c.c_uint32,
        c.c_uint32, c.c_uint32, c.c_void_p, c.POINTER(c.c_int),
    ]
    lib.spectre_gpu_msm_g1_combine_windows.restype = c.c_int
    lib.spectre_gpu_msm_g1_combine_windows.argtypes = [
        c.c_void_p, c.c_uint32, c.c_void_p,
    ]
    lib.spectre_gpu_fr_gate_eval.restype = c.c_int
    lib.spectre_gpu_fr_gate_eval.argtypes = [
        c.c_void_p, c.c_int, c.POINTER(c.c_void_p), c.c_uint32, c.c_void_p,
        c.c_uint32, c.POINTER(c.c_uint32), c.c_uint32, c.c_uint64, c.c_uint32,
        c.c_void_p, c.c_void_p,
    ]
    lib.spectre_gpu_msm_g1_combine.restype = c.c_int
    lib.spectre_gpu_msm_g1_combine.argtypes = [c.c_void_p, c.c_uint32, c.c_void_p]
    lib.spectre_gpu_ntt_fr.restype = c.c_int
    lib.spectre_gpu_ntt_fr.argtypes = [
        c.c_void_p, c.c_void_p, c.c_uint32, c.c_void_p, c.c_int, c.c_void_p,
    ]
    lib.spectre_gpu_ntt_fr_device.restype = c.c_int
    lib.spectre_gpu_ntt_fr_device.argtypes = [
        c.c_void_p, c.c_int, c.c_void_p, c.c_uint32, c.c_void_p, c.c_int,
        c.c_void_p,
    ]
    lib.spectre_gpu_fr_vec_op.restype = c.c_int
    lib.spectre_gpu_fr_vec_op.argtypes = [
        c.c_void_p, c.c_int, c.c_int, c.c_void_p, c.c_void_p, c.c_void_p,
        c.c_void_p, c.c_uint64,
    ]
    lib.spectre_gpu_malloc.restype = c.c_int
    lib.spectre_gpu_malloc.argtypes = [c.c_void_p, c.c_int, c.c_size_t,
                                       c.POINTER(c.c_void_p)]
    lib.spectre_gpu_free.restype = c.c_int
    lib.spectre_gpu_free.argtypes = [c.c_void_p, c.c_int, c.c_void_p]
    lib.spectre_gpu_upload.restype = c.c_int
    lib.spectre_gpu_upload.argtypes = [c.c_void_p, c.c_int, c.c_void_p,
                                       c.c_void_p, c.c_size_t]
    lib.spectre_gpu_download.restype = c.c_int
    lib.spectre_gpu_download.argtypes = [c.c_void_p, c.c_int, c.c_void_p,
                                         c.c_void_p, c.c_size_t]
    lib.spectre_gpu_synchronize.restype = c.c_int
    lib.spectre_gpu_synchronize.argtypes = [c.c_void_p, c.c_int]
    _lib = lib
    return lib


def version() -> str:
    return load_library().spectre_gpu_version().decode()


def combine_window_partials(partials: bytes, nshards: int) -> bytes:
    """Host-only: concatenated rank-ordered DISJOINT window slices (shard i =
    NUM_WINDOWS/nshards consecutive windows) -> affine result."""
    lib = load_library()
    assert len(partials) == NUM_WINDOWS * 96
    buf = (ctypes.c_uint8 * len(partials)).from_buffer_copy(partials)
    out = (ctypes.c_uint8 * 64)()
    rc = lib.spectre_gpu_msm_g1_combine_windows(buf, nshards, out)
    if rc != 0:
        raise RuntimeError(f"combine_windows failed rc={rc}: "
                           f"{lib.spectre_gpu_last_error().decode()}")
    return bytes(out)


def combine_partials(partials: bytes, nshards: int) -> bytes:
    """Host-only: combine shard window partials (rank order) to affine."""
    lib = load_library()
    assert len(partials) == nshards * PARTIALS_BYTES
    buf = (ctypes.c_uint8 * len(partials)).from_buffer_copy(partials)
    out = (ctypes.c_uint8 * 64)()
    rc = lib.spectre_gpu_msm_g1_combine(buf, nshards, out)
    if rc != 0:
        raise RuntimeError(f"combine failed rc={rc}: "
                           f"{lib.spectre_gpu_last_error().decode()}")
    return bytes(out)


class SpectreGpu:
    """A context over one or more MI355X devices. Raises without a GPU."""

    def __init__(self, device_ids=None):
        self._lib = load_library()
        if device_ids is None:
            device_ids = [0]
        arr = (ctypes.c_int * len(device_ids))(*device_ids)
        self._ctx = self._lib.spectre_gpu_init(len(device_ids), arr)
        if not self._ctx:
            raise RuntimeError(
                "spectre_gpu_init failed (no AMD GPU?): "
                + self._lib.spectre_gpu_last_error().decode()
            )

    def close(self):
        if getattr(self, "_ctx", None):
            self._lib.spectre_gpu_destroy(self._ctx)
            self._ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _check(self, rc, what):
        if rc != 0:
            raise RuntimeError(
                f"{what} failed rc={rc}: "
                f"{self._lib.spectre_gpu_last_error().decode()}"
            )

    # ---- host-pointer API (the halo2 seam) ----
    def msm(self, bases: bytes | None, scalars: bytes, n: int,
            canonical: bool = True, num_gpus: int = 1,
            bases_id: int = 0) -> bytes:
        out = (ctypes.c_uint8 * 64)()
        # read-only args pass zero-copy (the C side never writes them)
        b = ctypes.cast(ctypes.c_char_p(bases), ctypes.c_void_p) if bases else None
        s = ctypes.cast(ctypes.c_char_p(scalars), ctypes.c_void_p)
        rc = self._lib.spectre_gpu_msm_g1(
            self._ctx, bases_id, b, s, n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, num_gpus,
            out)
        self._check(rc, "msm_g1")
        return bytes(out)

    def msm_batch(self, bases: bytes | None, scalars: bytes, nbatch: int,
                  n: int, canonical: bool = True, bases_id: int = 0) -> list[bytes]:
        """nbatch MSMs over one shared base set (batch-major scalars)."""
        assert len(scalars) == nbatch * n * 32
        out = (ctypes.c_uint8 * (64 * nbatch))()
        b = ctypes.cast(ctypes.c_char_p(bases), ctypes.c_void_p) if bases else None
        s = ctypes.cast(ctypes.c_char_p(scalars), ctypes.c_void_p)
        rc = self._lib.spectre_gpu_msm_g1_batch(
            self._ctx, bases_id, b, s, nbatch, n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out)
        self._check(rc, "msm_g1_batch")
        raw = bytes(out)
        return [raw[64 * i:64 * (i + 1)] for i in range(nbatch)]

    def msm_batch_device(self, d_bases: int, d_scalars: int, nbatch: int,
                         n: int, canonical: bool = True,
                         dev: int = 0) -> list[bytes]:
        out = (ctypes.c_uint8 * (64 * nbatch))()
        rc = self._lib.spectre_gpu_msm_g1_batch_device(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), nbatch, n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out)
        self._check(rc, "msm_g1_batch_device")
        raw = bytes(out)
        return [raw[64 * i:64 * (i + 1)] for i in range(nbatch)]

    def ntt(self, data: bytes, log_n: int, omega: bytes, inverse: bool = False,
            coset_gen: bytes | None = None) -> bytes:
        assert log_n > 28 or len(data) == 32 << log_n, "data length != 32*2^log_n"
        buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
        om = (ctypes.c_uint8 * 32).from_buffer_copy(omega)
        cg = ((ctypes.c_uint8 * 32).from_buffer_copy(coset_gen)
              if coset_gen else None)
        rc = self._lib.spectre_gpu_ntt_fr(self._ctx, buf, log_n, om,
                                          1 if inverse else 0, cg)
        self._check(rc, "ntt_fr")
        return bytes(buf)

    # ---- device-resident API (device pointers, e.g. torch .data_ptr()) ----
    def msm_device(self, d_bases: int, d_scalars: int, n: int,
                   canonical: bool = True, dev: int = 0) -> bytes:
        out = (ctypes.c_uint8 * 64)()
        rc = self._lib.spectre_gpu_msm_g1_device(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out)
        self._check(rc, "msm_g1_device")
        return bytes(out)

    def msm_shard_device(self, d_bases: int, d_scalars: int, n: int,
                         canonical: bool = True, dev: int = 0) -> bytes:
        out = (ctypes.c_uint8 * PARTIALS_BYTES)()
        rc = self._lib.spectre_gpu_msm_g1_shard_device(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out)
        self._check(rc, "msm_g1_shard_device")
        return bytes(out)

    def msm_shard_device_async(self, d_bases: int, d_scalars: int, n: int,
                               canonical: bool = True, dev: int = 0):
        """Enqueue a full shard MSM on one of two per-device pipeline slots
        and return (partials_buffer, slot) WITHOUT synchronizing. Call
        msm_slot_wait(slot) before reading the buffer; keep the returned
        ctypes buffer referenced until then (async D2H writes into it)."""
        out = (ctypes.c_uint8 * PARTIALS_BYTES)()
        slot = ctypes.c_int(-1)
        rc = self._lib.spectre_gpu_msm_g1_shard_device_async(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out,
            ctypes.byref(slot))
        self._check(rc, "msm_g1_shard_device_async")
        return out, slot.value

    def msm_slot_wait(self, slot: int, dev: int = 0) -> None:
        rc = self._lib.spectre_gpu_msm_slot_wait(self._ctx, dev, slot)
        self._check(rc, "msm_slot_wait")

    def msm_shard_windows_device(self, d_bases: int, d_scalars: int, n: int,
                                 w_lo: int, w_cnt: int,
                                 canonical: bool = True,
                                 dev: int = 0) -> bytes:
        """Window-sharded shard: this rank's w_cnt windows over ALL n
        points (disjoint across ranks; exchange = pure allgather)."""
        out = (ctypes.c_uint8 * (96 * w_cnt))()
        rc = self._lib.spectre_gpu_msm_g1_shard_windows_device(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY,
            w_lo, w_cnt, out)
        self._check(rc, "msm_g1_shard_windows_device")
        return bytes(out)

    def msm_shard_windows_device_async(self, d_bases: int, d_scalars: int,
                                       n: int, w_lo: int, w_cnt: int,
                                       canonical: bool = True, dev: int = 0):
        out = (ctypes.c_uint8 * (96 * w_cnt))()
        slot = ctypes.c_int(-1)
        rc = self._lib.spectre_gpu_msm_g1_shard_windows_device_async(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY,
            w_lo, w_cnt, out, ctypes.byref(slot))
        self._check(rc, "msm_g1_shard_windows_device_async")
        return out, slot.value

    def msm_shard_device_timed(self, d_bases: int, d_scalars: int, n: int,
                               canonical: bool = True, dev: int = 0):
        """Returns (partials_bytes, stage_ms dict) — stage timings from HIP
        events on the library stream (see spectre_gpu.h)."""
        out = (ctypes.c_uint8 * PARTIALS_BYTES)()
        ms = (ctypes.c_double * 8)()
        rc = self._lib.spectre_gpu_msm_g1_shard_device_timed(
            self._ctx, dev, ctypes.c_void_p(d_bases),
            ctypes.c_void_p(d_scalars), n,
            SCALARS_CANONICAL if canonical else SCALARS_MONTGOMERY, out, ms)
        self._check(rc, "msm_g1_shard_device_timed")
        stages = dict(zip(
            ["digits", "sort", "offsets", "bucket_acc", "chunks", "reduce",
             "total", "real_entries"], list(ms)))
        return bytes(out), stages

    def ntt_device(self, d_data: int, log_n: int, omega: bytes,
                   inverse: bool = False, coset_gen: bytes | None = None,
                   dev: int = 0) -> None:
        om = (ctypes.c_uint8 * 32).from_buffer_copy(omega)
        cg = ((ctypes.c_uint8 * 32).from_buffer_copy(coset_gen)
              if coset_gen else None)
        rc = self._lib.spectre_gpu_ntt_fr_device(
            self._ctx, dev, ctypes.c_void_p(d_data), log_n, om,
            1 if inverse else 0, cg)
        self._check(rc, "ntt_fr_device")

    # gate-expression evaluator opcodes (spectre_gpu.h)
    GATE_COL, GATE_CONST, GATE_ADD, GATE_SUB, GATE_MUL, GATE_NEG = range(6)

    def gate_eval(self, d_cols: list[int], constants: bytes,
                  program: list[tuple[int, int, int]], n: int,
                  rot_scale: int = 1, y: bytes | None = None,
                  d_out: int = 0, dev: int = 0) -> None:
        """Evaluate a gate expression over n rows into device buffer d_out
        (out = out*y + v when y is given). program = [(op, a, rot)]."""
        cols = (ctypes.c_void_p * len(d_cols))(*d_cols)
        prog = (ctypes.c_uint32 * (3 * len(program)))()
        for i, (op, a, b) in enumerate(program):
            prog[3 * i] = op
            prog[3 * i + 1] = a
            prog[3 * i + 2] = b & 0xFFFFFFFF
        con = ((ctypes.c_uint8 * len(constants)).from_buffer_copy(constants)
               if constants else None)
        yb = (ctypes.c_uint8 * 32).from_buffer_copy(y) if y else None
        rc = self._lib.spectre_gpu_fr_gate_eval(
            self._ctx, dev, cols, len(d_cols), con,
            len(constants) // 32, prog, len(program), n, rot_scale, yb,
            ctypes.c_void_p(d_out))
        self._check(rc, "fr_gate_eval")

    VEC_ADD, VEC_SUB, VEC_MUL, VEC_SCALE, VEC_ADD_SCALED = range(5)

    def fr_vec_op(self, op: int, d_a: int, d_b: int | None, c: bytes | None,
                  d_out: int, n: int, dev: int = 0) -> None:
        """Pointwise Fr vector op on device buffers (Montgomery form)."""
        cc = (ctypes.c_uint8 * 32).from_buffer_copy(c) if c else None
        rc = self._lib.spectre_gpu_fr_vec_op(
            self._ctx, dev, op, ctypes.c_void_p(d_a),
            ctypes.c_void_p(d_b) if d_b else None, cc,
            ctypes.c_void_p(d_out), n)
        self._check(rc, "fr_vec_op")

    # ---- device memory helpers ----
    def malloc(self, nbytes: int, dev: int = 0) -> int:
        p = ctypes.c_void_p()
        self._check(self._lib.spectre_gpu_malloc(self._ctx, dev, nbytes,
                                                 ctypes.byref(p)), "malloc")
        return p.value

    def free(self, d_ptr: int, dev: int = 0) -> None:
        self._check(self._lib.spectre_gpu_free(self._ctx, dev,
                                               ctypes.c_void_p(d_ptr)), "free")

    def upload(self, d_dst: int, data: bytes, dev: int = 0) -> None:
        buf = ctypes.cast(ctypes.c_char_p(data), ctypes.c_void_p)
        self._check(self._lib.spectre_gpu_upload(self._ctx, dev,
                                                 ctypes.c_void_p(d_dst), buf,
                                                 len(data)), "upload")

    def download(self, d_src: int, nbytes: int, dev: int = 0) -> bytes:
        buf = (ctypes.c_uint8 * nbytes)()
        self._check(self._lib.spectre_gpu_download(self._ctx, dev, buf,
                                                   ctypes.c_void_p(d_src),
                                                   nbytes), "download")
        return bytes(buf)

    def synchronize(self, dev: int = 0) -> None:
        self._check(self._lib.spectre_gpu_synchronize(self._ctx, dev), "sync")

"""ctypes wrapper over oracle/liboracle.so — TEST INFRASTRUCTURE ONLY.

Importable only from tests/, __graft_entry__.smoke() (as the checker) and
bench.py's cpu_baseline leg. The product package (spectre_amd) never imports
this module.
"""
import ctypes
import os
import subprocess

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "liboracle.so")


def build(force: bool = False) -> None:
    if force or not os.path.exists(_LIB_PATH):
        subprocess.run(["make", "-C", _HERE], check=True, capture_output=True)


_lib = None


def lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_LIB_PATH)
        _lib.oracle_num_threads.restype = ctypes.c_int
        _lib.oracle_g1_is_on_curve.restype = ctypes.c_int
    return _lib


def _buf(b: bytes) -> ctypes.Array:
    return (ctypes.c_uint8 * len(b)).from_buffer_copy(b)


def _call2(name: str, a: bytes, b: bytes, outlen: int) -> bytes:
    out = (ctypes.c_uint8 * outlen)()
    getattr(lib(), name)(_buf(a), _buf(b), out)
    return bytes(out)


def _call1(name: str, a: bytes, outlen: int) -> bytes:
    out = (ctypes.c_uint8 * outlen)()
    getattr(lib(), name)(_buf(a), out)
    return bytes(out)


def fr_add(a, b): return _call2("oracle_fr_add", a, b, 32)
def fr_sub(a, b): return _call2("oracle_fr_sub", a, b, 32)
def fr_mul(a, b): return _call2("oracle_fr_mul", a, b, 32)
def fr_inv(a): return _call1("oracle_fr_inv", a, 32)
def fr_pow(a, e): return _call2("oracle_fr_pow", a, e, 32)
def fr_to_canonical(a): return _call1("oracle_fr_to_canonical", a, 32)
def fr_from_canonical(a): return _call1("oracle_fr_from_canonical", a, 32)
def fq_add(a, b): return _call2("oracle_fq_add", a, b, 32)
def fq_sub(a, b): return _call2("oracle_fq_sub", a, b, 32)
def fq_mul(a, b): return _call2("oracle_fq_mul", a, b, 32)
def fq_inv(a): return _call1("oracle_fq_inv", a, 32)
def g1_add(a, b): return _call2("oracle_g1_add", a, b, 64)
def g1_neg(a): return _call1("oracle_g1_neg", a, 64)
def g1_mul(p, k): return _call2("oracle_g1_mul", p, k, 64)


def g2_add(a, b): return _call2("oracle_g2_add", a, b, 128)
def g2_neg(a): return _call1("oracle_g2_neg", a, 128)
def g2_mul(p, k): return _call2("oracle_g2_mul", p, k, 128)


def g2_is_on_curve(p: bytes) -> bool:
    l = lib()
    l.oracle_g2_is_on_curve.restype = ctypes.c_int
    return bool(l.oracle_g2_is_on_curve(_buf(p)))


def msm_g2(bases: bytes, scalars_canon: bytes, n: int) -> bytes:
    out = (ctypes.c_uint8 * 128)()
    lib().oracle_msm_g2(_buf(bases), _buf(scalars_canon), ctypes.c_uint64(n), out)
    return bytes(out)


def g1_is_on_curve(p: bytes) -> bool:
    return bool(lib().oracle_g1_is_on_curve(_buf(p)))


def msm(bases: bytes, scalars: bytes, n: int, scalars_canonical: bool = True) -> bytes:
    assert len(bases) == 64 * n and len(scalars) == 32 * n
    out = (ctypes.c_uint8 * 64)()
    lib().oracle_msm_g1(_buf(bases), _buf(scalars), ctypes.c_uint64(n),
                        ctypes.c_int(1 if scalars_canonical else 0), out)
    return bytes(out)


def ntt(data: bytes, log_n: int, omega: bytes, inverse: bool = False,
        coset_gen: bytes | None = None) -> bytes:
    n = 1 << log_n
    assert len(data) == 32 * n
    buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
    lib().oracle_ntt_fr(buf, ctypes.c_uint32(log_n), _buf(omega),
                        ctypes.c_int(1 if inverse else 0),
                        _buf(coset_gen) if coset_gen else None)
    return bytes(buf)


def gen_msm_inputs(n: int, seed: int, fast: bool = False) -> tuple[bytes, bytes]:
    scalars = (ctypes.c_uint8 * (32 * n))()
    bases = (ctypes.c_uint8 * (64 * n))()
    fn = lib().oracle_gen_msm_inputs_fast if fast else lib().oracle_gen_msm_inputs
    fn(ctypes.c_uint64(n), ctypes.c_uint64(seed), scalars, bases)
    return bytes(scalars), bytes(bases)


def gen_fr_vector(n: int, seed: int) -> bytes:
    out = (ctypes.c_uint8 * (32 * n))()
    lib().oracle_gen_fr_vector(ctypes.c_uint64(n), ctypes.c_uint64(seed), out)
    return bytes(out)


def num_threads() -> int:
    return lib().oracle_num_threads()

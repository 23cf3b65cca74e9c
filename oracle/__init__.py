"""CPU oracle bindings — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this package, and only as the parity checker / CPU baseline — never
as the product path (DESIGN.md "Oracle discipline").
"""
import ctypes
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libbn254_oracle.so")


def _build():
    subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


if not os.path.exists(_SO):
    _build()

_lib = ctypes.CDLL(_SO)

_lib.oracle_g1_add.restype = ctypes.c_int
for _f in ("bls_g2_add", "bls_g2_mul", "bls_g2_msm", "bls_g2_gen_points"):
    getattr(_lib, f"oracle_{_f}").restype = ctypes.c_int
_lib.oracle_g1_mul.restype = ctypes.c_int
_lib.oracle_g1_msm.restype = ctypes.c_int
_lib.oracle_g1_msm_naive.restype = ctypes.c_int
_lib.oracle_g1_msm_jacobian.restype = ctypes.c_int
_lib.oracle_g1_combine_jacobian.restype = ctypes.c_int
_lib.oracle_fr_ntt.restype = ctypes.c_int
_lib.oracle_fr_dft_naive.restype = ctypes.c_int
_lib.oracle_gen_points.restype = ctypes.c_int
_lib.oracle_num_threads.restype = ctypes.c_int


def _buf(b):
    return (ctypes.c_uint8 * len(b)).from_buffer_copy(b)


def g1_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.oracle_g1_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def g1_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.oracle_g1_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def g1_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.oracle_g1_msm(_buf(points), _buf(scalars), ctypes.c_size_t(n), out)
    return rc, bytes(out)


def g1_msm_naive(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.oracle_g1_msm_naive(_buf(points), _buf(scalars), ctypes.c_size_t(n), out)
    return rc, bytes(out)


def g1_msm_jacobian(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_g1_msm_jacobian(_buf(points), _buf(scalars), ctypes.c_size_t(n), out)
    return rc, bytes(out)


def g1_combine_jacobian(jacobians: bytes, g: int):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.oracle_g1_combine_jacobian(_buf(jacobians), ctypes.c_size_t(g), out)
    return rc, bytes(out)


def fr_ntt(elems: bytes, n: int, inverse: bool):
    buf = _buf(elems)
    rc = _lib.oracle_fr_ntt(buf, ctypes.c_size_t(n), ctypes.c_int(1 if inverse else 0))
    return rc, bytes(buf)


def fr_dft_naive(elems: bytes, n: int, inverse: bool):
    out = (ctypes.c_uint8 * (32 * n))()
    rc = _lib.oracle_fr_dft_naive(_buf(elems), out, ctypes.c_size_t(n),
                                  ctypes.c_int(1 if inverse else 0))
    return rc, bytes(out)


def gen_fr(seed: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (32 * n))()
    _lib.oracle_gen_fr(ctypes.c_uint64(seed), ctypes.c_size_t(n), out)
    return bytes(out)


def gen_points(start: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (64 * n))()
    rc = _lib.oracle_gen_points(ctypes.c_uint64(start), ctypes.c_size_t(n), out)
    assert rc == 0
    return bytes(out)


def fq_mulmod(a: bytes, b: bytes) -> bytes:
    out = (ctypes.c_uint8 * 32)()
    _lib.oracle_fq_mulmod(_buf(a), _buf(b), out)
    return bytes(out)


def fr_mulmod(a: bytes, b: bytes) -> bytes:
    out = (ctypes.c_uint8 * 32)()
    _lib.oracle_fr_mulmod(_buf(a), _buf(b), out)
    return bytes(out)


def num_threads() -> int:
    return _lib.oracle_num_threads()


# ---- BLS12-381 G1 (semantics: bls_blst.rs) ----

_lib.oracle_bls_g1_add.restype = ctypes.c_int
_lib.oracle_bls_g1_mul.restype = ctypes.c_int
_lib.oracle_bls_g1_msm.restype = ctypes.c_int
_lib.oracle_bls_g1_msm_naive.restype = ctypes.c_int
_lib.oracle_bls_g1_msm_jacobian.restype = ctypes.c_int
_lib.oracle_bls_g1_combine_jacobian.restype = ctypes.c_int
_lib.oracle_bls_gen_points.restype = ctypes.c_int


def bls_g2_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.oracle_bls_g2_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def bls_g2_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.oracle_bls_g2_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def bls_g2_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.oracle_bls_g2_msm(_buf(points), _buf(scalars),
                                ctypes.c_size_t(n), out)
    return rc, bytes(out)


def bls_g2_gen_points(start: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (192 * n))()
    rc = _lib.oracle_bls_g2_gen_points(ctypes.c_uint64(start),
                                       ctypes.c_size_t(n), out)
    assert rc == 0
    return bytes(out)


def bls_g1_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_bls_g1_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def bls_g1_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_bls_g1_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def bls_g1_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_bls_g1_msm(_buf(points), _buf(scalars), ctypes.c_size_t(n), out)
    return rc, bytes(out)


def bls_g1_msm_naive(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_bls_g1_msm_naive(_buf(points), _buf(scalars),
                                      ctypes.c_size_t(n), out)
    return rc, bytes(out)


def bls_g1_msm_jacobian(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 144)()
    rc = _lib.oracle_bls_g1_msm_jacobian(_buf(points), _buf(scalars),
                                         ctypes.c_size_t(n), out)
    return rc, bytes(out)


def bls_g1_combine_jacobian(jacobians: bytes, g: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.oracle_bls_g1_combine_jacobian(_buf(jacobians),
                                             ctypes.c_size_t(g), out)
    return rc, bytes(out)


def bls_gen_points(start: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (96 * n))()
    rc = _lib.oracle_bls_gen_points(ctypes.c_uint64(start), ctypes.c_size_t(n), out)
    assert rc == 0
    return bytes(out)


def bls_gen_fr(seed: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (32 * n))()
    _lib.oracle_bls_gen_fr(ctypes.c_uint64(seed), ctypes.c_size_t(n), out)
    return bytes(out)

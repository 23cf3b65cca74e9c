"""ctypes binding to libethrex_mi355.so (the gfx950 HIP core).

Loads the in-tree shared library built by __graft_entry__.build().  Import
fails loudly if the library is absent: the product path never falls back
to CPU.
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libethrex_mi355.so")

if not os.path.exists(_SO):
    raise ImportError(
        f"HIP core library missing: {_SO}. Run __graft_entry__.build() "
        "(hipcc --offload-arch=gfx950). The product path has no CPU fallback.")

_lib = ctypes.CDLL(_SO)

EM_OK = 0
EM_ERR_POINT = 1
EM_ERR_INPUT = 2
EM_ERR_HIP = 3

_lib.ethrex_mi355_version.restype = ctypes.c_char_p
_lib.ethrex_mi355_last_error.restype = ctypes.c_char_p
for _f in ("device_count", "set_device", "bn254_g1_add", "bn254_g1_mul",
           "bn254_g1_msm", "bn254_fr_ntt", "bn254_g1_combine",
           "msm_plan_create", "msm_plan_destroy", "msm_upload_points",
           "msm_gen_points", "msm_download_points", "msm_upload_scalars",
           "msm_run", "msm_run_partial", "msm_run_async", "msm_sync",
           "msm_run_partial_async", "msm_wait_one", "bn254_g1_combine_cpu",
           "msm_last_times", "msm_combine", "msm_scalars_from_ntt",
           "ntt_device_data",
           "ntt_plan_create", "ntt_plan_destroy", "ntt_upload", "ntt_run",
           "ntt_download", "ntt_last_times"):
    getattr(_lib, f"ethrex_mi355_{_f}").restype = ctypes.c_int


class HipCoreError(RuntimeError):
    def __init__(self, rc, what):
        self.rc = rc
        super().__init__(f"{what}: rc={rc} ({last_error()})")


def _check(rc, what):
    if rc != EM_OK:
        raise HipCoreError(rc, what)


def _buf(b):
    return (ctypes.c_uint8 * len(b)).from_buffer_copy(b)


def version() -> str:
    return _lib.ethrex_mi355_version().decode()


def last_error() -> str:
    return (_lib.ethrex_mi355_last_error() or b"").decode()


def device_count() -> int:
    n = ctypes.c_int(0)
    _lib.ethrex_mi355_device_count(ctypes.byref(n))
    return n.value


def set_device(d: int):
    _check(_lib.ethrex_mi355_set_device(ctypes.c_int(d)), "set_device")


def g1_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.ethrex_mi355_bn254_g1_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def g1_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.ethrex_mi355_bn254_g1_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def g1_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.ethrex_mi355_bn254_g1_msm(_buf(points), _buf(scalars),
                                        ctypes.c_size_t(n), out)
    return rc, bytes(out)


def fr_ntt(elems: bytes, n: int, inverse: bool):
    buf = _buf(elems)
    rc = _lib.ethrex_mi355_bn254_fr_ntt(buf, ctypes.c_size_t(n),
                                        ctypes.c_int(1 if inverse else 0))
    return rc, bytes(buf)


def g1_combine(jacobians: bytes, count: int):
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.ethrex_mi355_bn254_g1_combine(_buf(jacobians),
                                            ctypes.c_size_t(count), out)
    return rc, bytes(out)


def g1_combine_cpu(jacobians: bytes, count: int) -> bytes:
    """Host-side combine of the N>1 exchange payload (world-size 96-B
    Jacobian partials) — boundary glue; keeps GPU streams untouched while
    pipelined steps are in flight."""
    out = (ctypes.c_uint8 * 64)()
    rc = _lib.ethrex_mi355_bn254_g1_combine_cpu(_buf(jacobians),
                                                ctypes.c_size_t(count), out)
    if rc != EM_OK:
        raise HipCoreError(rc, "bn254_g1_combine_cpu")
    return bytes(out)


def gen_fr(seed: int, n: int) -> bytes:
    """Deterministic Fr elements (host-side; BASELINE.md input scheme)."""
    out = (ctypes.c_uint8 * (32 * n))()
    _lib.ethrex_mi355_gen_fr(ctypes.c_uint64(seed), ctypes.c_size_t(n), out)
    return bytes(out)


class MsmPlan:
    """Device-resident MSM plan: upload once, run many (bench/pipeline)."""

    def __init__(self, n: int):
        self.n = n
        self._p = ctypes.c_void_p()
        self._pending = []
        _check(_lib.ethrex_mi355_msm_plan_create(ctypes.c_size_t(n),
                                                 ctypes.byref(self._p)),
               "msm_plan_create")

    def upload_points(self, points: bytes):
        _check(_lib.ethrex_mi355_msm_upload_points(self._p, _buf(points)),
               "msm_upload_points")

    def gen_points(self, start: int = 0):
        _check(_lib.ethrex_mi355_msm_gen_points(self._p, ctypes.c_uint64(start)),
               "msm_gen_points")

    def download_points(self) -> bytes:
        out = (ctypes.c_uint8 * (64 * self.n))()
        _check(_lib.ethrex_mi355_msm_download_points(self._p, out),
               "msm_download_points")
        return bytes(out)

    def upload_scalars(self, scalars: bytes):
        _check(_lib.ethrex_mi355_msm_upload_scalars(self._p, _buf(scalars)),
               "msm_upload_scalars")

    def run(self) -> bytes:
        out = (ctypes.c_uint8 * 64)()
        _check(_lib.ethrex_mi355_msm_run(self._p, out), "msm_run")
        return bytes(out)

    def run_partial(self) -> bytes:
        out = (ctypes.c_uint8 * 96)()
        _check(_lib.ethrex_mi355_msm_run_partial(self._p, out), "msm_run_partial")
        return bytes(out)

    def last_times(self):
        t = (ctypes.c_double * 5)()
        _check(_lib.ethrex_mi355_msm_last_times(self._p, t), "msm_last_times")
        return {"digits_sort_ms": t[0], "bucket_acc_ms": t[1],
                "reduce_ms": t[2], "combine_ms": t[3], "total_ms": t[4]}

    def run_async(self):
        """Enqueue one pipelined MSM step (sort chain of the next step
        overlaps this step's compute chain).  Result delivered at sync()."""
        buf = (ctypes.c_uint8 * 64)()
        _check(_lib.ethrex_mi355_msm_run_async(self._p, buf), "msm_run_async")
        self._pending.append(buf)

    def run_partial_async(self):
        """Pipelined shard step: delivers the 96-B Jacobian partial at
        sync() (the N>1 exchange payload)."""
        buf = (ctypes.c_uint8 * 96)()
        _check(_lib.ethrex_mi355_msm_run_partial_async(self._p, buf),
               "msm_run_partial_async")
        self._pending.append(buf)

    def sync(self) -> bytes:
        """Drain the pipeline; returns the LAST pipelined result."""
        _check(_lib.ethrex_mi355_msm_sync(self._p), "msm_sync")
        outs = [bytes(b) for b in self._pending]
        self._pending = []
        return outs[-1] if outs else b""

    def wait_one(self) -> bytes:
        """Deliver ONLY the oldest pending pipelined step (later steps keep
        running on the GPU).  The N>1 loop exchanges step k's partial while
        the GPU computes step k+1."""
        if not self._pending:
            return b""
        _check(_lib.ethrex_mi355_msm_wait_one(self._p), "msm_wait_one")
        return bytes(self._pending.pop(0))

    def scalars_from_ntt(self, ntt_plan, offset: int = 0):
        """wrap-pipeline handoff: take this plan's n scalars from the NTT
        plan's device-resident output at element `offset` (on-device, no
        PCIe; the sp1.rs:122-134 wrap flow feeds the witness NTT output
        into the proving MSM)."""
        ptr = ctypes.c_void_p()
        nn = ctypes.c_size_t(0)
        _check(_lib.ethrex_mi355_ntt_device_data(ntt_plan._p,
                                                 ctypes.byref(ptr),
                                                 ctypes.byref(nn)),
               "ntt_device_data")
        if offset + self.n > nn.value:
            raise HipCoreError(EM_ERR_INPUT,
                               f"scalars_from_ntt: offset {offset} + n "
                               f"{self.n} exceeds NTT size {nn.value}")
        _check(_lib.ethrex_mi355_msm_scalars_from_ntt(
            self._p, ptr, ctypes.c_uint64(offset)), "msm_scalars_from_ntt")

    def combine(self, jacobians: bytes, count: int) -> bytes:
        """combine Jacobian partials reusing this plan's device buffers"""
        out = (ctypes.c_uint8 * 64)()
        _check(_lib.ethrex_mi355_msm_combine(self._p, _buf(jacobians),
                                             ctypes.c_size_t(count), out),
               "msm_combine")
        return bytes(out)

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_msm_plan_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


class NttPlan:
    def __init__(self, n: int):
        self.n = n
        self._p = ctypes.c_void_p()
        _check(_lib.ethrex_mi355_ntt_plan_create(ctypes.c_size_t(n),
                                                 ctypes.byref(self._p)),
               "ntt_plan_create")

    def upload(self, elems: bytes):
        _check(_lib.ethrex_mi355_ntt_upload(self._p, _buf(elems)), "ntt_upload")

    def run(self, inverse: bool = False):
        _check(_lib.ethrex_mi355_ntt_run(self._p,
                                         ctypes.c_int(1 if inverse else 0)),
               "ntt_run")

    def download(self) -> bytes:
        out = (ctypes.c_uint8 * (32 * self.n))()
        _check(_lib.ethrex_mi355_ntt_download(self._p, out), "ntt_download")
        return bytes(out)

    def last_times(self):
        t = (ctypes.c_double * 3)()
        _check(_lib.ethrex_mi355_ntt_last_times(self._p, t), "ntt_last_times")
        return {"bitrev_ms": t[0], "stages_ms": t[1], "total_ms": t[2]}

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_ntt_plan_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


# ==== BLS12-381 G1 (bls_blst.rs semantics) ====

for _f in ("bls12381_g1_add", "bls12381_g1_mul", "bls12381_g1_msm",
           "bls12381_g1_combine", "bls_msm_plan_create", "bls_msm_plan_destroy",
           "bls_msm_upload_points", "bls_msm_gen_points",
           "bls_msm_download_points", "bls_msm_upload_scalars", "bls_msm_run",
           "bls_msm_run_async", "bls_msm_sync", "bls12381_g2_add",
           "bls12381_g2_mul", "bls12381_g2_msm", "bls_g2_msm_plan_create",
           "bls_g2_msm_plan_destroy", "bls_g2_msm_upload_points",
           "bls_g2_msm_gen_points", "bls_g2_msm_download_points",
           "bls_g2_msm_upload_scalars", "bls_g2_msm_run",
           "bls_g2_msm_run_async", "bls_g2_msm_sync",
           "bls_g2_msm_run_partial", "bls_g2_msm_last_times",
           "keccak256_batch", "keccak_plan_create", "keccak_plan_destroy",
           "keccak_upload", "keccak_run", "keccak_download",
           "keccak_last_ms", "mpt_create", "mpt_destroy", "mpt_max_depth",
           "mpt_level_encode", "mpt_level_set_hashes", "mpt_root",
           "bls_msm_run_partial", "bls_msm_last_times", "bls_msm_precompute"):
    getattr(_lib, f"ethrex_mi355_{_f}").restype = ctypes.c_int


def bls_g1_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.ethrex_mi355_bls12381_g1_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def bls_g1_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.ethrex_mi355_bls12381_g1_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def bls_g1_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.ethrex_mi355_bls12381_g1_msm(_buf(points), _buf(scalars),
                                           ctypes.c_size_t(n), out)
    return rc, bytes(out)


def bls_g1_combine(jacobians: bytes, count: int):
    out = (ctypes.c_uint8 * 96)()
    rc = _lib.ethrex_mi355_bls12381_g1_combine(_buf(jacobians),
                                               ctypes.c_size_t(count), out)
    return rc, bytes(out)


def bls_gen_fr(seed: int, n: int) -> bytes:
    out = (ctypes.c_uint8 * (32 * n))()
    _lib.ethrex_mi355_bls_gen_fr(ctypes.c_uint64(seed), ctypes.c_size_t(n), out)
    return bytes(out)


class BlsMsmPlan:
    """Device-resident BLS12-381 G1 MSM plan (blob-KZG commitment shape)."""

    def __init__(self, n: int):
        self.n = n
        self._p = ctypes.c_void_p()
        self._pending = []
        _check(_lib.ethrex_mi355_bls_msm_plan_create(ctypes.c_size_t(n),
                                                     ctypes.byref(self._p)),
               "bls_msm_plan_create")

    def upload_points(self, points: bytes):
        _check(_lib.ethrex_mi355_bls_msm_upload_points(self._p, _buf(points)),
               "bls_msm_upload_points")

    def gen_points(self, start: int = 0):
        _check(_lib.ethrex_mi355_bls_msm_gen_points(self._p,
                                                    ctypes.c_uint64(start)),
               "bls_msm_gen_points")

    def download_points(self) -> bytes:
        out = (ctypes.c_uint8 * (96 * self.n))()
        _check(_lib.ethrex_mi355_bls_msm_download_points(self._p, out),
               "bls_msm_download_points")
        return bytes(out)

    def upload_scalars(self, scalars: bytes):
        _check(_lib.ethrex_mi355_bls_msm_upload_scalars(self._p, _buf(scalars)),
               "bls_msm_upload_scalars")

    def precompute(self):
        """fixed-base table (setup points fixed across blobs)"""
        _check(_lib.ethrex_mi355_bls_msm_precompute(self._p),
               "bls_msm_precompute")

    def run(self) -> bytes:
        out = (ctypes.c_uint8 * 96)()
        _check(_lib.ethrex_mi355_bls_msm_run(self._p, out), "bls_msm_run")
        return bytes(out)

    def run_partial(self) -> bytes:
        out = (ctypes.c_uint8 * 144)()
        _check(_lib.ethrex_mi355_bls_msm_run_partial(self._p, out),
               "bls_msm_run_partial")
        return bytes(out)

    def run_async(self):
        buf = (ctypes.c_uint8 * 96)()
        _check(_lib.ethrex_mi355_bls_msm_run_async(self._p, buf),
               "bls_msm_run_async")
        self._pending.append(buf)

    def sync(self) -> bytes:
        _check(_lib.ethrex_mi355_bls_msm_sync(self._p), "bls_msm_sync")
        outs = [bytes(b) for b in self._pending]
        self._pending = []
        return outs[-1] if outs else b""

    def last_times(self):
        t = (ctypes.c_double * 5)()
        _check(_lib.ethrex_mi355_bls_msm_last_times(self._p, t),
               "bls_msm_last_times")
        return {"digits_sort_ms": t[0], "bucket_acc_ms": t[1],
                "reduce_ms": t[2], "combine_ms": t[3], "total_ms": t[4]}

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_bls_msm_plan_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


def bls_g2_add(p1: bytes, p2: bytes):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.ethrex_mi355_bls12381_g2_add(_buf(p1), _buf(p2), out)
    return rc, bytes(out)


def bls_g2_mul(point: bytes, scalar: bytes):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.ethrex_mi355_bls12381_g2_mul(_buf(point), _buf(scalar), out)
    return rc, bytes(out)


def bls_g2_msm(points: bytes, scalars: bytes, n: int):
    out = (ctypes.c_uint8 * 192)()
    rc = _lib.ethrex_mi355_bls12381_g2_msm(_buf(points), _buf(scalars),
                                           ctypes.c_size_t(n), out)
    return rc, bytes(out)


class BlsG2MsmPlan:
    """Device-resident BLS12-381 G2 MSM plan (EIP-2537 G2MSM shape)."""

    def __init__(self, n: int):
        self.n = n
        self._p = ctypes.c_void_p()
        self._pending = []
        _check(_lib.ethrex_mi355_bls_g2_msm_plan_create(ctypes.c_size_t(n),
                                                        ctypes.byref(self._p)),
               "bls_g2_msm_plan_create")

    def upload_points(self, points: bytes):
        _check(_lib.ethrex_mi355_bls_g2_msm_upload_points(self._p,
                                                          _buf(points)),
               "bls_g2_msm_upload_points")

    def gen_points(self, start: int = 0):
        _check(_lib.ethrex_mi355_bls_g2_msm_gen_points(
            self._p, ctypes.c_uint64(start)), "bls_g2_msm_gen_points")

    def download_points(self) -> bytes:
        out = (ctypes.c_uint8 * (192 * self.n))()
        _check(_lib.ethrex_mi355_bls_g2_msm_download_points(self._p, out),
               "bls_g2_msm_download_points")
        return bytes(out)

    def upload_scalars(self, scalars: bytes):
        _check(_lib.ethrex_mi355_bls_g2_msm_upload_scalars(self._p,
                                                           _buf(scalars)),
               "bls_g2_msm_upload_scalars")

    def run(self) -> bytes:
        out = (ctypes.c_uint8 * 192)()
        _check(_lib.ethrex_mi355_bls_g2_msm_run(self._p, out),
               "bls_g2_msm_run")
        return bytes(out)

    def run_partial(self) -> bytes:
        out = (ctypes.c_uint8 * 288)()
        _check(_lib.ethrex_mi355_bls_g2_msm_run_partial(self._p, out),
               "bls_g2_msm_run_partial")
        return bytes(out)

    def run_async(self):
        buf = (ctypes.c_uint8 * 192)()
        _check(_lib.ethrex_mi355_bls_g2_msm_run_async(self._p, buf),
               "bls_g2_msm_run_async")
        self._pending.append(buf)

    def sync(self) -> bytes:
        _check(_lib.ethrex_mi355_bls_g2_msm_sync(self._p), "bls_g2_msm_sync")
        outs = [bytes(b) for b in self._pending]
        self._pending = []
        return outs[-1] if outs else b""

    def last_times(self):
        t = (ctypes.c_double * 5)()
        _check(_lib.ethrex_mi355_bls_g2_msm_last_times(self._p, t),
               "bls_g2_msm_last_times")
        return {"digits_sort_ms": t[0], "bucket_acc_ms": t[1],
                "reduce_ms": t[2], "combine_ms": t[3], "total_ms": t[4]}

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_bls_g2_msm_plan_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


def keccak256_batch(msgs: bytes, offsets, n: int):
    """Batched Ethereum keccak256 (one-shot, PCIe-inclusive); offsets is a
    list of n+1 byte offsets into msgs."""
    offs = (ctypes.c_uint64 * (n + 1))(*offsets)
    out = (ctypes.c_uint8 * (32 * n))()
    rc = _lib.ethrex_mi355_keccak256_batch(
        _buf(msgs) if msgs else None, offs, ctypes.c_size_t(n), out)
    return rc, bytes(out)


class KeccakPlan:
    """Device-resident batched keccak256 (witness/trie hashing shape)."""

    def __init__(self, max_bytes: int, max_n: int):
        self._p = ctypes.c_void_p()
        _check(_lib.ethrex_mi355_keccak_plan_create(
            ctypes.c_size_t(max_bytes), ctypes.c_size_t(max_n),
            ctypes.byref(self._p)), "keccak_plan_create")

    def upload(self, msgs: bytes, offsets):
        self.n = len(offsets) - 1
        offs = (ctypes.c_uint64 * len(offsets))(*offsets)
        _check(_lib.ethrex_mi355_keccak_upload(
            self._p, _buf(msgs) if msgs else None, offs,
            ctypes.c_size_t(self.n)), "keccak_upload")

    def run(self):
        _check(_lib.ethrex_mi355_keccak_run(self._p), "keccak_run")

    def download(self) -> bytes:
        out = (ctypes.c_uint8 * (32 * self.n))()
        _check(_lib.ethrex_mi355_keccak_download(self._p, out),
               "keccak_download")
        return bytes(out)

    def last_ms(self) -> float:
        ms = ctypes.c_double(0)
        _check(_lib.ethrex_mi355_keccak_last_ms(self._p, ctypes.byref(ms)),
               "keccak_last_ms")
        return ms.value

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_keccak_plan_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


class MptBuilder:
    """Native MPT structure builder (host C) with caller-driven per-level
    hashing — §8f row 4 witness-generation speedup.  Fixed 32-byte sorted
    distinct keys (hashed-key state/storage trie shape); see
    include/ethrex_mi355.h and ethrex_amd/trie.py (trie_root_native)."""

    def __init__(self, keys32: bytes, vals: bytes, val_offs):
        self.n = len(val_offs) - 1
        self._p = ctypes.c_void_p()
        offs = (ctypes.c_uint64 * len(val_offs))(*val_offs)
        _check(_lib.ethrex_mi355_mpt_create(
            _buf(keys32), _buf(vals) if vals else (ctypes.c_uint8 * 1)(),
            offs, ctypes.c_size_t(self.n), ctypes.byref(self._p)),
            "mpt_create")
        # reusable level-encode buffers (branch worst case ~580 B/node)
        self._cap = 600 * (2 * self.n + 4) + 4096
        self._buf = (ctypes.c_uint8 * self._cap)()
        self._offs = (ctypes.c_uint64 * (2 * self.n + 4))()

    def max_depth(self) -> int:
        d = ctypes.c_int(0)
        _check(_lib.ethrex_mi355_mpt_max_depth(self._p, ctypes.byref(d)),
               "mpt_max_depth")
        return d.value

    def level_encode(self, depth: int):
        """-> (msg bytes, offsets list) for this level's to-hash nodes"""
        k = ctypes.c_size_t(0)
        _check(_lib.ethrex_mi355_mpt_level_encode(
            self._p, ctypes.c_int(depth), self._buf, self._offs,
            ctypes.c_size_t(self._cap), ctypes.c_size_t(2 * self.n + 3),
            ctypes.byref(k)), "mpt_level_encode")
        nh = k.value
        offs = [self._offs[i] for i in range(nh + 1)]
        return bytes(self._buf[:offs[-1]]), offs

    def level_set_hashes(self, depth: int, hashes: bytes):
        _check(_lib.ethrex_mi355_mpt_level_set_hashes(
            self._p, ctypes.c_int(depth),
            _buf(hashes) if hashes else (ctypes.c_uint8 * 1)(),
            ctypes.c_size_t(len(hashes) // 32)), "mpt_level_set_hashes")

    def root(self) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        _check(_lib.ethrex_mi355_mpt_root(self._p, out), "mpt_root")
        return bytes(out)

    def destroy(self):
        if self._p:
            _lib.ethrex_mi355_mpt_destroy(self._p)
            self._p = ctypes.c_void_p()

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass

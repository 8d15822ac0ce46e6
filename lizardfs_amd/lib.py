"""ctypes binding over liblizec.so (see include/lizec.h for the contract)."""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "liblizec.so")


class LizecError(RuntimeError):
    pass


_ERRNAMES = {
    -1: "LIZEC_ESINGULAR (decode matrix not invertible)",
    -2: "LIZEC_EINVAL",
    -3: "LIZEC_ENOGPU (no MI355X visible; the product path has no CPU fallback)",
    -4: "LIZEC_EHIP",
    -5: "LIZEC_ENOMEM",
}

_lib = None


def lib():
    """Load liblizec.so.  Raises loudly if the HIP extension is missing."""
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        raise LizecError(
            f"liblizec.so not found at {_SO}; build it with "
            f"`make -C lizardfs_amd/csrc` (or __graft_entry__.build())")
    L = ctypes.CDLL(_SO)

    u8p = ctypes.POINTER(ctypes.c_uint8)
    L.gf_gen_rs_matrix.argtypes = [u8p, ctypes.c_int, ctypes.c_int]
    L.gf_gen_cauchy1_matrix.argtypes = [u8p, ctypes.c_int, ctypes.c_int]
    L.gf_invert_matrix.restype = ctypes.c_int
    L.gf_invert_matrix.argtypes = [u8p, u8p, ctypes.c_int]
    L.ec_init_tables.argtypes = [ctypes.c_int, ctypes.c_int, u8p, u8p]
    L.ec_encode_data.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int,
                                 u8p, ctypes.POINTER(ctypes.c_void_p),
                                 ctypes.POINTER(ctypes.c_void_p)]
    L.lizec_crc32.restype = ctypes.c_uint32
    L.lizec_crc32.argtypes = [ctypes.c_uint32, ctypes.c_char_p, ctypes.c_uint32]
    L.lizec_crc32_combine.restype = ctypes.c_uint32
    L.lizec_crc32_combine.argtypes = [ctypes.c_uint32, ctypes.c_uint32,
                                      ctypes.c_uint32]
    L.lizec_rs_tables.restype = ctypes.c_int
    L.lizec_rs_tables.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
                                  ctypes.c_uint64, ctypes.c_uint64, u8p,
                                  ctypes.POINTER(ctypes.c_int),
                                  ctypes.POINTER(ctypes.c_int)]
    L.lizec_rs_encode_tables.restype = ctypes.c_int
    L.lizec_rs_encode_tables.argtypes = [ctypes.c_int, ctypes.c_int, u8p]
    for f in ("lizec_slice_type_ec", "lizec_slice_is_ec",
              "lizec_slice_data_parts", "lizec_slice_parity_parts",
              "lizec_chunk_part_id", "lizec_chunk_part_slice_type",
              "lizec_chunk_part_index"):
        getattr(L, f).restype = ctypes.c_int
    L.lizec_chunk_part_length.restype = ctypes.c_int64
    L.lizec_chunk_part_length.argtypes = [ctypes.c_int, ctypes.c_int,
                                          ctypes.c_int64]
    L.lizec_gpu_count.restype = ctypes.c_int
    L.lizec_engine_create.restype = ctypes.c_int
    L.lizec_engine_create.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                                      ctypes.c_int]
    L.lizec_engine_destroy.argtypes = [ctypes.c_void_p]
    L.lizec_engine_sync.restype = ctypes.c_int
    L.lizec_engine_sync.argtypes = [ctypes.c_void_p]
    L.lizec_ec_encode_batch.restype = ctypes.c_int
    L.lizec_ec_encode_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.c_int, u8p,
        ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_uint64),
        ctypes.c_int, ctypes.c_void_p]
    L.lizec_ec_plan_create.restype = ctypes.c_int
    L.lizec_ec_plan_create.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.c_int, u8p,
        ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_uint64),
        ctypes.c_int, ctypes.POINTER(ctypes.c_void_p)]
    L.lizec_ec_plan_run.restype = ctypes.c_int
    L.lizec_ec_plan_run.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    L.lizec_ec_plan_destroy.argtypes = [ctypes.c_void_p]
    L.lizec_crc32_batch.restype = ctypes.c_int
    L.lizec_crc32_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint32, ctypes.c_uint64,
        ctypes.c_uint32, ctypes.c_void_p, ctypes.c_void_p]
    L.lizec_scrub_batch.restype = ctypes.c_int
    L.lizec_scrub_batch.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint64),
        ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32), ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p]
    L.lizec_scrub_batch_strided.restype = ctypes.c_int
    L.lizec_scrub_batch_strided.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint64),
        ctypes.POINTER(ctypes.c_uint32), ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32), ctypes.c_int, ctypes.c_uint32,
        ctypes.c_uint32, ctypes.c_void_p, ctypes.c_void_p]
    L.lizec_host_alloc.restype = ctypes.c_int
    L.lizec_host_alloc.argtypes = [ctypes.POINTER(ctypes.c_void_p),
                                   ctypes.c_uint64]
    L.lizec_host_free.argtypes = [ctypes.c_void_p]
    L.lizec_replicate_run.restype = ctypes.c_int
    L.lizec_replicate_run.argtypes = [
        ctypes.c_void_p, ctypes.c_uint64, ctypes.c_int, ctypes.c_int, u8p,
        ctypes.POINTER(ctypes.c_uint64), u8p, ctypes.c_uint32,
        ctypes.c_uint32, ctypes.c_uint32, ctypes.POINTER(ctypes.c_uint64),
        ctypes.c_int, ctypes.c_int]
    _lib = L
    return _lib


def check(code, what=""):
    if code != 0:
        raise LizecError(f"{what or 'lizec call'} failed: "
                         f"{_ERRNAMES.get(code, code)}")
    return code


def gpu_count():
    return lib().lizec_gpu_count()


def pinned_empty(shape, dtype=None):
    """numpy array backed by pinned (hipHostMalloc) memory — gives the
    streaming pipeline (lizec_replicate_run) true async copies.  The
    allocation is freed when the returned array is garbage-collected."""
    import numpy as np
    dtype = np.dtype(dtype or np.uint8)
    n = int(np.prod(shape)) * dtype.itemsize
    p = ctypes.c_void_p()
    check(lib().lizec_host_alloc(ctypes.byref(p), max(n, 1)),
          "lizec_host_alloc")
    buf = (ctypes.c_uint8 * n).from_address(p.value)
    arr = np.frombuffer(buf, dtype=dtype).reshape(shape)
    # tie the allocation's lifetime to the array
    arr.base.base._lizec_finalizer = _HostMem(p)  # type: ignore[attr-defined]
    return arr


class _HostMem:
    def __init__(self, p):
        self._p = p

    def __del__(self):
        try:
            lib().lizec_host_free(self._p)
        except Exception:
            pass


_engines = {}


def engine(device=0):
    """Per-device engine handle (one internal stream + scratch)."""
    e = _engines.get(device)
    if e is None:
        h = ctypes.c_void_p()
        check(lib().lizec_engine_create(ctypes.byref(h), device),
              "lizec_engine_create")
        _engines[device] = e = h
    return e

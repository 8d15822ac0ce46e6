"""XOR slice family (goals xor2..xor9) — SURVEY §8f row 1.

Reference semantics: a xor-level-L slice has 1 parity part (slice part
index 0, slice_traits.h:98 kXorParityPart) and L data parts (slice parts
1..L, data index = part-1, slice_traits.h:283-288).  Parity = byte-XOR of
the L data parts (xor_read_plan.h:39 RecoverParity, chunk_writer.cc:373-381
xor branch of computeParityBlock, common/block_xor.cc:47 blockXor); a
missing data part = parity XOR the other data parts.

XOR over GF(2^8) is the EC kernel with all coefficients = 1, so both
operations run on the same HIP path (ec_encode_kernel with a ones-table);
no separate kernel needed and the parity tests pin it against the oracle's
RS machinery and plain numpy XOR.
"""
import ctypes

import numpy as np
import torch

from . import lib as L
from . import slice_traits


class XorSlice:
    """Batched xor-level-N parity/recovery on one GPU."""

    def __init__(self, level, device=0):
        if not (2 <= level <= 9):
            raise ValueError("xor level must be in [2, 9] (goal.h:110-117)")
        self.level = level
        self.device = device
        self.slice_type = slice_traits.K_XOR2 + level - 2
        self._engine = L.engine(device)
        self._lib = L.lib()
        self._tables = {}

    def _ones_tables(self, srcs):
        t = self._tables.get(srcs)
        if t is None:
            ones = np.ones(srcs, np.uint8)
            t = np.zeros(32 * srcs, np.uint8)
            self._lib.ec_init_tables(
                srcs, 1, ones.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
                t.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
            self._tables[srcs] = t
        return t

    def _xor_reduce(self, parts, out):
        """out[s] = XOR of parts[j][s] over j (device, via the EC kernel)."""
        S, plen = parts[0].shape[0], parts[0].shape[-1]
        if (out.dtype != torch.uint8 or not out.is_cuda
                or not out.is_contiguous() or tuple(out.shape) != (S, plen)):
            raise ValueError(f"out must be a contiguous CUDA uint8 "
                             f"[S={S}, L={plen}] tensor")
        srcs = len(parts)
        tbl = self._ones_tables(srcs)
        rows = np.arange(S, dtype=np.uint64)
        src = np.empty((S, srcs), np.uint64)
        for j, t in enumerate(parts):
            if t.dtype != torch.uint8 or not t.is_cuda or t.stride(-1) != 1:
                raise ValueError("parts must be CUDA uint8 with contiguous rows")
            if tuple(t.shape) != (S, plen):
                raise ValueError(f"part {j} is {tuple(t.shape)}, "
                                 f"expected [{S}, {plen}]")
            src[:, j] = t.data_ptr() + rows * np.uint64(t.stride(0))
        dst = (out.data_ptr() + rows * np.uint64(plen)).astype(np.uint64)
        stream = torch.cuda.current_stream(self.device).cuda_stream
        L.check(self._lib.lizec_ec_encode_batch(
            self._engine, plen, srcs, 1,
            tbl.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            np.ascontiguousarray(src.ravel()).ctypes.data_as(
                ctypes.POINTER(ctypes.c_uint64)),
            np.ascontiguousarray(dst).ctypes.data_as(
                ctypes.POINTER(ctypes.c_uint64)),
            S, ctypes.c_void_p(stream)), "lizec_ec_encode_batch(xor)")
        return out

    def parity_batch(self, data, out=None):
        """data: uint8 CUDA [S, level, L] -> parity [S, L]
        (XorReadPlan::RecoverParity semantics, batched)."""
        S, lvl, plen = data.shape
        if lvl != self.level:
            raise ValueError(f"expected {self.level} data parts")
        if plen % 16:
            raise ValueError("part length must be a multiple of 16")
        if out is None:
            out = torch.empty((S, plen), dtype=torch.uint8, device=data.device)
        return self._xor_reduce([data[:, j, :] for j in range(lvl)], out)

    def recover_data_batch(self, parity, other_data, out=None):
        """Missing data part = parity XOR the other level-1 data parts.
        parity: [S, L]; other_data: list of level-1 [S, L] fragments."""
        if len(other_data) != self.level - 1:
            raise ValueError(f"need {self.level - 1} surviving data parts")
        S, plen = parity.shape
        if out is None:
            out = torch.empty((S, plen), dtype=torch.uint8,
                              device=parity.device)
        return self._xor_reduce([parity] + list(other_data), out)

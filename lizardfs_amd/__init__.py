"""lizardfs_amd — MI355X-native erasure-coding engine for the LizardFS
chunkserver EC hot path.

The compute path is liblizec.so (HIP/CDNA4 kernels behind the C ABI declared
in include/lizec.h, the reference's own ISA-L-shaped plugin seam).  This
package is the host-side mirror of the reference's EC surfaces:

  lizardfs_amd.slice_traits — Goal::Slice::Type / ChunkPartType algebra
  lizardfs_amd.ec           — batched ReedSolomon encode/decode on GPU
  lizardfs_amd.crc          — per-block CRC32 (GPU batch + host scalar)

PyTorch supplies device memory and streams only; all byte-level compute is
the HIP library.  There is NO CPU fallback on the product path: using the
GPU APIs without liblizec.so or without an MI355X raises immediately.
"""
from . import slice_traits  # noqa: F401
from .lib import LizecError, gpu_count  # noqa: F401
from .lib import lib as load_lib  # noqa: F401
from . import ec  # noqa: F401
from . import crc  # noqa: F401

__all__ = ["slice_traits", "ec", "crc", "load_lib", "gpu_count", "LizecError"]

"""hipHostRegister/hipHostUnregister via ctypes.

Flash checkpoint's D2H drain wants the POSIX shm mapping page-locked so
`hipMemcpyAsync` runs at full PCIe gen5 rate and truly async (ref premise:
flash_checkpoint.md:68-71 quotes ~32 GB/s PCIe4; MI355X is gen5 ~63 GB/s).
torch can't pin foreign memory, so we register the mapping directly with the
HIP runtime. Falls back cleanly when no GPU / no libamdhip64 is present.
"""

import ctypes
import ctypes.util
from typing import Optional

from dlrover_amd.common.log import logger

_lib: Optional[ctypes.CDLL] = None
_checked = False


def _hip() -> Optional[ctypes.CDLL]:
    global _lib, _checked
    if _checked:
        return _lib
    _checked = True
    for name in ("libamdhip64.so", "libamdhip64.so.7", "libamdhip64.so.6"):
        try:
            _lib = ctypes.CDLL(name)
            break
        except OSError:
            continue
    if _lib is not None:
        _lib.hipHostRegister.argtypes = [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_uint]
        _lib.hipHostRegister.restype = ctypes.c_int
        _lib.hipHostUnregister.argtypes = [ctypes.c_void_p]
        _lib.hipHostUnregister.restype = ctypes.c_int
    return _lib


def host_register(ptr: int, nbytes: int) -> bool:
    """Page-lock [ptr, ptr+nbytes). Returns True on success."""
    lib = _hip()
    if lib is None or nbytes == 0:
        return False
    rc = lib.hipHostRegister(ctypes.c_void_p(ptr), ctypes.c_size_t(nbytes), 0)
    if rc != 0:
        logger.warning("hipHostRegister(%d bytes) failed rc=%d — D2H will be pageable", nbytes, rc)
        return False
    return True


def host_unregister(ptr: int) -> None:
    lib = _hip()
    if lib is not None:
        lib.hipHostUnregister(ctypes.c_void_p(ptr))

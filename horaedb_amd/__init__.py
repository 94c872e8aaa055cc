# horaedb_amd — MI355X-native implementation of HoraeDB's metric-engine
# scan/aggregate hot path. This package is a thin ctypes wrapper
# over the C-ABI drop-in boundary (include/horaedb_hx.h, libhoraedb_hx.so:
# C++ host + gfx950 HIP kernels). The compute path is GPU-only: if the
# library is missing, import fails loudly; if no GPU is visible, scan calls
# raise HxError(HX_ERR_NO_GPU). There is no CPU fallback.
from .store import (  # noqa: F401
    HxError,
    Store,
    AGG_SUM,
    AGG_COUNT,
    AGG_MIN,
    AGG_MAX,
    AGG_AVG,
    lib_path,
)

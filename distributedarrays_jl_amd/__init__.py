"""distributedarrays_jl_amd — MI355X-native DArray local-compute path.

A from-scratch re-implementation of the hot path of
JuliaParallel/DistributedArrays.jl (BASELINE.json `north_star`): the
DArray/distribute/localpart/map/reduce/broadcast/matmul surface of
src/darray.jl stays (exports mirror src/DistributedArrays.jl:14-21), but
each rank's localpart lives in HBM on one MI355X and the per-worker hot
loops of src/mapreduce.jl + src/broadcast.jl + src/linalg.jl are
hand-written gfx950 HIP kernels behind the C ABI in include/darray_hip.h,
with cross-worker aggregation as RCCL over xGMI.

The HIP extension is the ONLY compute path — importing this package
without libdarray_hip.so raises, and every op fails loudly if the GPU is
absent.  The CPU restatement used by tests lives in oracle/ and is never
imported here.
"""

from ._ffi import DArrayError
from . import comm, geometry, plan, spmd, expr
from .darray import (DArray, dzeros, dones, dfill, drand, drandn,
                     distribute, localpart, localindices, d_closeall,
                     bytes_in_use, ddata, dgather, locate, allowscalar,
                     dfromfunction)
from .ops import (map_, dmap, map2_, elementwise, map2_scalar_, elementwise_scalar, broadcast_fma, axpy_,
                  add_, scale_, mapreduce, dsum, dprod, dmaximum, dminimum,
                  dextrema, dmean, dcount, dall, dany, ddot, dnorm, dmatmul, dreduce_dims,
                  dsum_dims, dprod_dims, dmaximum_dims, dminimum_dims,
                  dmean_dims, dmatvec, dmatvec_adj, gather_box, map_general,
                  map_localparts, map_localparts_, redistribute,
                  dmapslices, dppeval, dcast, dreshape,
                  map2_general,
                  broadcast_fma_general, dsort, dtranspose, ddiag_lmul, ddiag_rmul,
                  dgetindex, dmul_)

__all__ = [
    "DArray", "DArrayError", "comm", "geometry", "plan", "spmd", "expr",
    "dzeros", "dones", "dfill", "drand", "drandn", "distribute",
    "localpart", "localindices", "d_closeall", "bytes_in_use",
    "ddata", "dgather", "locate", "allowscalar", "dfromfunction",
    "map_", "dmap", "map2_", "elementwise", "map2_scalar_", "elementwise_scalar", "broadcast_fma", "axpy_",
    "add_", "scale_", "mapreduce", "dsum", "dprod", "dmaximum",
    "dminimum", "dextrema", "dmean", "dcount", "dall", "dany", "ddot", "dnorm", "dmatmul",
    "dreduce_dims", "dsum_dims", "dprod_dims", "dmaximum_dims",
    "dminimum_dims", "dmean_dims", "dmatvec", "dmatvec_adj",
    "map_localparts", "map_localparts_", "redistribute",
    "dmapslices", "dppeval", "dcast", "dreshape",
    "map2_general",
    "gather_box", "map_general", "broadcast_fma_general", "dsort",
    "dtranspose", "ddiag_lmul", "ddiag_rmul", "dgetindex", "dmul_",
]

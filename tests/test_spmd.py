"""SPMD veneer (spmd.jl analog over RCCL).  World-1 semantics here;
the underlying grouped send/recv transport is exercised by the gloo
schedule tests and the RCCL probe (tools/rccl_2proc_probe.py)."""
import numpy as np
import pytest

from oracle import philox


@pytest.mark.gpu
def test_spmd_world1_semantics():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    sp = dja.spmd
    sp.barrier()   # no-op at world 1
    a = philox.fill_uniform_f64(1000, 3)
    assert np.array_equal(sp.bcast_host(a, root=0), a)
    parts = [a.reshape(10, 100)]
    got = sp.scatter_host(parts, root=0)
    assert np.array_equal(got, parts[0])
    g = sp.gather_host(a, root=0)
    assert len(g) == 1 and np.array_equal(g[0], a)


def test_spmd_imports():
    import distributedarrays_jl_amd as dja
    for name in ("barrier", "sendto", "recvfrom", "sendrecv", "bcast",
                 "bcast_host", "scatter_host", "gather_host"):
        assert hasattr(dja.spmd, name)

"""Rank bootstrap: one OS process per GPU (1 reference worker <-> 1 rank,
SURVEY.md §5).  Reads torchrun's RANK/WORLD_SIZE/LOCAL_RANK environment;
the RCCL uniqueId rendezvous is a shared-filesystem file (single node)."""
import os
import atexit

from . import _ffi
from ._ffi import check

_state = {"inited": False}


def default_uid_path():
    tag = "%s_%s" % (os.environ.get("MASTER_PORT", "0"), os.getppid())
    return os.path.join(os.environ.get("TMPDIR", "/tmp"),
                        "darray_rccl_uid_%s" % tag)


def init(device=None, rank=None, nranks=None, uid_path=None):
    """Idempotent; collective when nranks > 1."""
    if _state["inited"]:
        return rank_info()
    # dmabuf IPC is the only mode the host driver supports (env note in
    # the build environment); harmless if already exported
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if rank is None:
        rank = int(os.environ.get("RANK", "0"))
    if nranks is None:
        nranks = int(os.environ.get("WORLD_SIZE", "1"))
    if device is None:
        device = int(os.environ.get("LOCAL_RANK", str(rank)))
    if uid_path is None:
        uid_path = os.environ.get("DA_UID_PATH", default_uid_path())
    if nranks > 1 and rank == 0 and os.path.exists(uid_path):
        os.unlink(uid_path)   # stale rendezvous from a crashed run
    check(_ffi.lib.da_init(device, rank, nranks,
                           uid_path.encode() if nranks > 1 else None))
    _state.update(inited=True, rank=rank, nranks=nranks, device=device)
    atexit.register(shutdown)
    return rank_info()


def shutdown():
    if _state["inited"]:
        _ffi.lib.da_shutdown()
        _state["inited"] = False


def rank_info():
    return _state.get("rank", 0), _state.get("nranks", 1)


def initialized():
    return _state["inited"]


def rank():
    return _state.get("rank", 0)


def nranks():
    return _state.get("nranks", 1)

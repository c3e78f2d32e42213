"""axon — an MI355X-native ring-allreduce data-parallel training engine.

A from-scratch framework with the capability set and `hvd.*` API of the
reference engine: `import axon as hvd` and use
hvd.init() / hvd.DistributedOptimizer() / hvd.broadcast_parameters() as
usual. The data plane is RCCL over xGMI with hand-written CDNA4 HIP
kernels; coordination is a single-node unix-socket control plane; one
process per MI355X GPU.
"""
import torch as _torch  # load libtorch before the extension  # noqa: F401

__version__ = "0.1.0"

from axon.basics import (  # noqa: F401,E402
    cross_rank,
    cross_size,
    init,
    is_initialized,
    local_rank,
    local_size,
    rank,
    shutdown,
    size,
    start_timeline,
    stop_timeline,
)
from axon.compression import Compression  # noqa: F401,E402
from axon.functions import (  # noqa: F401,E402
    allgather_object,
    broadcast_object,
    broadcast_optimizer_state,
    broadcast_parameters,
)
from axon.mpi_ops import (  # noqa: F401,E402
    Adasum,
    Average,
    Max,
    Min,
    Product,
    ReduceOp,
    Sum,
    allgather,
    allgather_async,
    allreduce,
    allreduce_,
    allreduce_async,
    allreduce_async_,
    alltoall,
    alltoall_async,
    barrier,
    broadcast,
    broadcast_,
    broadcast_async,
    broadcast_async_,
    grouped_allreduce,
    grouped_allreduce_,
    grouped_allreduce_async,
    grouped_allreduce_async_,
    join,
    poll,
    reducescatter,
    reducescatter_async,
    synchronize,
)
from axon.optimizer import DistributedOptimizer  # noqa: F401,E402
from axon.process_sets import (  # noqa: F401,E402
    ProcessSet,
    add_process_set,
    global_process_set,
    process_set_ranks,
    remove_process_set,
)
from axon import elastic  # noqa: F401,E402  (hvd.elastic.run / State)


def nccl_built():
    """Horovod-compat introspection: the GPU data plane (RCCL) is always
    compiled in."""
    return True


def mpi_built():
    return False


def gloo_built():
    return False


def cuda_built():
    """True in the horovod introspection sense of 'GPU support built' —
    the backend is HIP/RCCL on ROCm."""
    return True


def rocm_built():
    return True


def mpi_threads_supported():
    """hvd-compat: no MPI in this engine (socket control plane)."""
    return False


def mpi_enabled():
    return False


def gloo_enabled():
    return False


def is_homogeneous():
    """Single-node engine: every rank shares the host, so always True."""
    return True

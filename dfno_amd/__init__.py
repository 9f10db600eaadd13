"""dfno_amd — MI355X-native distributed Fourier Neural Operator framework.

A from-scratch rebuild of the capabilities of slimgroup/dfno
(reference mounted at /root/reference) for AMD Instinct MI355X (gfx950):
PyTorch-ROCm + hand-written HIP/CDNA4 kernels for the fused hot ops + RCCL
collectives over xGMI (one process per GPU, ``torch.distributed`` backend
"nccl" == RCCL) instead of the reference's DistDL/MPI stack.

Public API mirrors the reference package's star-exports
(/root/reference/dfno/__init__.py): models, losses, partition utilities,
normalisation helpers — plus the native comm primitives.
"""

from .partition import (
    Partition,
    init_distributed,
    finalize_distributed,
    is_distributed,
    zero_volume_tensor,
    compute_distribution_info,
    create_root_partition,
    create_standard_partitions,
)
from .comm import (
    Broadcast,
    SumReduce,
    Repartition,
    DistributedTranspose,
    AllReduceSum,
    ZeroVolumeCorrectorFunction,
)
from .nn import (
    BroadcastedLinear,
    DistributedFNOBlock,
    DistributedFNONd,
    DistributedFNO,
    DistributedRelativeLpLoss,
    DistributedMSELoss,
    DistributedBatchNorm,
)
from .utils import (
    alphabet,
    generate_batch_indices,
    get_env,
    get_gpu_memory,
    profile_gpu_memory,
    unit_guassian_normalize,
    unit_gaussian_normalize,
    unit_gaussian_denormalize,
)

__version__ = "0.1.0"

from .checkpoint import reshard_checkpoint  # noqa: E402

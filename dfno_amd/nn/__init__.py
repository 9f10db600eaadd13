from .linear import BroadcastedLinear
from .block import DistributedFNOBlock
from .fno import DistributedFNONd, DistributedFNO
from .loss import DistributedRelativeLpLoss, DistributedMSELoss
from .batchnorm import DistributedBatchNorm

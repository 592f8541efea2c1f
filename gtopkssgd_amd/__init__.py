"""gtopkssgd_amd — MI355X-native gTop-k sparsified synchronous SGD framework.

A from-scratch rebuild of the capabilities of hclhkbu/gtopkssgd (gTop-k
S-SGD, ICDCS 2019, arXiv:1901.04359) designed for AMD Instinct MI355X
(gfx950 / CDNA4): PyTorch-ROCm compute, hand-written HIP kernels for the
compression hot path (error-feedback accumulate, top-k select, sparse
merge, fused SGD), and RCCL point-to-point collectives over xGMI for the
recursive-halving global top-k merge.

Public API (mirrors the reference's DistributedOptimizer-style surface,
see SURVEY.md L4 `upstream:distributed_optimizer.py`):

    import gtopkssgd_amd as gt
    compressor = gt.compressors['gtopk'](density=0.001)
    opt = gt.DistributedOptimizer(torch.optim.SGD(...), named_parameters,
                                  compression=compressor, density=0.001)
    ...
    loss.backward(); opt.step()
"""

__version__ = "0.1.0"

from .compression import (  # noqa: F401
    NoneCompressor,
    TopKCompressor,
    compressors,
)
from .parallel.optimizer import DistributedOptimizer, rank, size  # noqa: F401
from .parallel.allreducer import (  # noqa: F401
    gtopk_sparse_allreduce,
    topk_sparse_allreduce,
    dense_allreduce,
)
from .ops.fusedbn import FusedBNAct  # noqa: F401

"""ddp_tricks_amd — an MI355X-native DDP training-tricks framework.

A from-scratch re-design of the capabilities of
Lance0218/Pytorch-DistributedDataParallel-Training-Tricks for AMD MI355X
(gfx950, CDNA4): PyTorch-ROCm front-end, hand-written HIP kernels for the
compute path (MFMA implicit-GEMM convs, fused BN/ReLU, fused CE, fused
Lookahead-SGD, amp unscale/inf-check), and RCCL-over-xGMI bucketed gradient
all-reduce overlapped with backward.

Public surface mirrors the reference package façade
(reference: utils/__init__.py:1-5):
    same_seeds, EarlyStopping, Lookahead, Toy_Net, iterate_loader, train
"""

__version__ = "0.1.0"

from .utils.callbacks import same_seeds, EarlyStopping  # noqa: F401
from .utils.lookahead import Lookahead                  # noqa: F401
from .models.toy_net import Toy_Net                     # noqa: F401
from .utils.engine import iterate_loader                # noqa: F401
from .utils.train import train                          # noqa: F401

__all__ = [
    "same_seeds",
    "EarlyStopping",
    "Lookahead",
    "Toy_Net",
    "iterate_loader",
    "train",
]

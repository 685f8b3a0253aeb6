"""pytorch_ps_mpi_amd — MI355X-native async parameter-server training engine.

Brand-new implementation of the capabilities of stsievert/pytorch_ps_mpi
(reference surveyed in SURVEY.md): a drop-in torch.optim.Optimizer with a
pluggable gradient-compression codec, replicated / sync-PS / AsySG-InCon
exchange modes — re-architected for one 8×MI355X node: flat device buffers,
RCCL over xGMI (torch.distributed "nccl"), hand-written CDNA4 HIP kernels
for every hot op (ops/csrc/ps_kernels.hip), fp32 master weights under bf16
compute.

Public API (reference parity: __init__.py:1 exported MPI_PS, Adam, SGD):

    from pytorch_ps_mpi_amd import PS, SGD, Adam
    opt = SGD(model.named_parameters(), lr=0.1, momentum=0.9,
              code="topk:0.01", mode="async")
    loss.backward(); loss, metrics = opt.step(loss=loss)
"""

from . import codecs, models, ops
from .codecs import (HostCodec, Identity, QuantInt8, TopK,
                     TopKThreshold, get_codec)
from .optim import PS, SGD, Adam
from .parallel.comm import Comm, init_distributed

def MPI_PS(named_params, *, optim="sgd", code=None, **kwargs):
    """Reference-compatible constructor (ps.py:53-59): selects the update
    rule via ``optim='sgd'|'adam'`` and takes the codec as ``code``."""
    cls = {"sgd": SGD, "adam": Adam}.get(optim)
    if cls is None:
        raise ValueError(f"optim must be 'sgd' or 'adam', got {optim!r}")
    return cls(named_params, code=code, **kwargs)

__all__ = [
    "PS", "MPI_PS", "SGD", "Adam",
    "Identity", "TopK", "TopKThreshold", "QuantInt8", "HostCodec",
    "get_codec",
    "Comm", "init_distributed",
    "codecs", "models", "ops",
]

__version__ = "0.1.0"

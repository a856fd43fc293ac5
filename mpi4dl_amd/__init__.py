"""mpi4dl_amd — MI355X-native distributed training framework for
very-high-resolution CNNs (spatial / layer / pipeline / GEMS / data
parallelism), built from scratch for one 8xMI355X node:

* PyTorch-ROCm at the framework layer (one process per GPU),
* RCCL over xGMI for all communication (``torch.distributed`` backend
  "nccl" is RCCL on ROCm): grouped point-to-point for halo exchange,
  pipeline edges and GEMS mirror traffic; bucketed collectives for
  gradient reduction,
* hand-written CDNA4 (gfx950) HIP kernels for the hot ops (halo
  pack/unpack, implicit-GEMM MFMA conv, BatchNorm, pooling) in
  ``mpi4dl_amd/csrc``.

Capability parity target: OSU-Nowlab/MPI4DL ("torchgems") v0.6 — see
SURVEY.md at the repo root for the full component inventory this package
implements. Public API mirrors the torchgems surface (model_generator,
train_model, train_model_spatial, train_model_master,
train_spatial_model_master, a Communicator in place of MPIComm) so a
reference user can switch without relearning the vocabulary.
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
from .comm import (  # noqa: F401
    Communicator,
    GradReducer,
    init_distributed,
    sync_comms_for_master,
)
from .parallel.partition import model_generator  # noqa: F401
from .parallel.pipeline import train_model  # noqa: F401

# Heavier engines are imported lazily by their modules:
#   mpi4dl_amd.parallel.spatial.train_model_spatial
#   mpi4dl_amd.parallel.gems.train_model_master
#   mpi4dl_amd.parallel.gems_spatial.train_spatial_model_master

MPIComm = Communicator  # torchgems-compat alias (reference: src/torchgems/comm.py:44)

"""parallel_cnn_amd — an MI355X-native CNN training framework.

Capability parity with the reference Tamerkobba/Parallel-CNN:
one framework, one backend — a CPU reference path (C++ / threaded, replaces
the reference's Sequential/ and Openmp/ variants), hand-written gfx950 HIP
kernels (replaces CUDA/), and RCCL data parallelism over xGMI (replaces MPI/).
"""

__version__ = "0.1.0"

from . import config  # noqa: F401

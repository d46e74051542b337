"""matrixone_amd — MI355X-native backend for MatrixOne's vector-index search
hot path (pkg/vectorindex IVF/HNSW ANN probe), built from scratch for CDNA4:
hand-written HIP kernels behind the reference's own cgo C ABI shape
(include/moann.h mirrors cgo/cuvs/ivf_flat_c.h), one process per GPU with
RCCL over xGMI for sharded search.

Product path only — the parity oracle lives in oracle/ and is never imported
here. The HIP extension is mandatory: every compute call fails loudly when
matrixone_amd/libmoann_hip.so is missing or no GPU is present.
"""

from . import engine  # noqa: F401
from .ivfflat import (IndexConfig, IndexSource, IvfflatSearch,  # noqa: F401
                      RuntimeConfig)
from .cache import Cache, VectorIndexCache  # noqa: F401

__version__ = "0.1.0"

"""poseidon_amd: an MI355X-native distributed CNN training framework with
PMLS-Caffe/Poseidon's capabilities (prototxt Net/Solver API, .caffemodel
checkpoints, DWBP-overlapped data parallelism, sufficient-factor broadcast)
built on hand-written HIP/CDNA4 kernels + RCCL over xGMI."""

__version__ = "0.1.0"

from .core.context import ctx, init
from .core.blob import Blob
from .core.net import Net, TRAIN, TEST
from . import layers  # registers all layer types
from .solver.solver import SGDSolver, NesterovSolver, AdaGradSolver, get_solver
from .proto import Message, parse_text, to_text

__all__ = ["ctx", "init", "Blob", "Net", "TRAIN", "TEST", "SGDSolver",
           "NesterovSolver", "AdaGradSolver", "get_solver", "Message",
           "parse_text", "to_text"]

"""MI355X-native distributed LLM training framework.

Brand-new framework with the capability surface of
aws-neuron/neuronx-distributed-training (the reference blueprint lives in
/root/repo/SURVEY.md), built MI355X-first: PyTorch-ROCm eager + hand-written
HIP/CDNA4 kernels for the hot path, RCCL collectives over xGMI, one process
per GPU.
"""

__version__ = "0.1.0"

from . import parallel  # noqa: F401

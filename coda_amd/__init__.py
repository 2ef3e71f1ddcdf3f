"""coda_amd: MI355X-native consensus-driven active model selection.

A from-scratch AMD-native engine with the capability set of CODA
(ICCV 2025): given an (H, N, C) tensor of H models' post-softmax
predictions over N unlabeled points, actively acquire oracle labels to
identify the best model. Compute core = PyTorch-ROCm + hand-written
gfx950 (CDNA4) HIP kernels; the model axis shards across GPUs with RCCL
collectives over xGMI.
"""
from .selectors import CODA
from .datasets import Dataset
from .oracle import Oracle
from .base import ModelSelector

__version__ = "0.1.0"
__all__ = ["CODA", "Dataset", "Oracle", "ModelSelector", "__version__"]

"""zaremba_amd — MI355X-native LSTM language-model training framework.

A from-scratch rebuild of the capabilities of ahmetumutdurmus/zaremba
(Zaremba, Sutskever & Vinyals, "Recurrent Neural Network Regularization",
arXiv:1409.2329) designed for AMD Instinct MI355X (gfx950, CDNA4):

  * hand-written HIP kernels (MFMA gate GEMMs, fused LSTM cell, philox
    dropout, fused log-softmax+NLL, fused grad-clip+SGD) — zaremba_amd/csrc/
  * PyTorch-ROCm as the tensor substrate; custom autograd.Functions route
    the hot path through the HIP kernels
  * data-parallel scaling over RCCL/xGMI (one process per GPU) with
    bucketed gradient all-reduce overlapped with the BPTT backward

Reference CLI surface (main.py / ensemble.py flags) is preserved; see
/root/reference/main.py:10-26 for the original flag set this mirrors.
"""

__version__ = "0.1.0"

from . import data  # noqa: F401
from .models.lstm_lm import Model  # noqa: F401

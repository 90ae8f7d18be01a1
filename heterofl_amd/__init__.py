"""heterofl_amd — an MI355X-native HeteroFL federated-learning engine.

A from-scratch re-design of
diaoenmao/HeteroFL-Computation-and-Communication-Efficient-Federated-Learning-
for-Heterogeneous-Clients for AMD Instinct MI355X (gfx950): PyTorch-ROCm +
hand-written HIP/CDNA4 kernels + RCCL over xGMI.  Same CLI surface,
control_name grammar and checkpoint layout as the reference; new engine:
clients batched into grouped kernels, hipGraph-captured local steps, padded
RCCL combine.
"""
__version__ = '0.1.0'

from .config import cfg, default_config, load_config
from .control import process_control, parse_control_name

__all__ = ['cfg', 'default_config', 'load_config', 'process_control',
           'parse_control_name', '__version__']

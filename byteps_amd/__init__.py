"""byteps_amd — an MI355X-native gradient-synchronization framework.

From-scratch rebuild of the capabilities of bytedance/byteps (push_pull,
DistributedOptimizer, DistributedDataParallel, hierarchical PS, gradient
compression, priority scheduling, tracing, elastic) designed for AMD
Instinct MI355X: PyTorch-ROCm + hand-written HIP/CDNA4 kernels + RCCL
over xGMI + a native C++ parameter server.  See DESIGN.md.
"""

__version__ = "0.1.0"

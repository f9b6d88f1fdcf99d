"""ROCnRDMA-AMD: MI355X-native GPU-direct RDMA framework.

Re-designed capabilities of rocmarchive/ROCnRDMA (the amdp2p PeerDirect
bridge) for gfx950: kernel bridge sources under module/, userspace
harness under harness/, and this Python package for kernels, transports,
sweeps and multi-GPU fan-out.
"""

__version__ = "0.1.0"

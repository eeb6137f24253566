"""brpc_amd: a brand-new MI355X-native RPC runtime with apache/brpc's
capabilities (Channel/Server/Controller API, IOBuf, fibers, protocols),
rebuilt for CDNA4: IOBuf blocks in HBM3E, gfx950 HIP kernels on the
byte-hot path, RCCL over xGMI for combo-channel fan-out."""
from brpc_amd._core import *  # noqa: F401,F403
from brpc_amd import _core as core  # noqa: F401

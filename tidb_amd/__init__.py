"""tidb_amd — MI355X-native vectorized executor for TiDB's analytical hot path.

The product engine lives in tidb_amd/csrc (HIP/CDNA4 kernels behind the
C-ABI in include/gx_executor.h). This package holds the Python plumbing:
ctypes bindings, plan construction, and chunk marshalling used by tests and
bench.py. The compute path is the native library — there is no Python or
eager-CPU fallback for it.
"""

"""spark_rapids_jni_amd — MI355X-native Spark columnar-execution support library.

A from-scratch AMD MI355X (CDNA4/gfx950) implementation of the capability
surface of NVIDIA/spark-rapids-jni: Spark-exact columnar
kernels written directly in HIP, an OOM-retry resource adaptor, the Kudo
shuffle serialization format, and RCCL-over-xGMI multi-GPU exchange.

Layering (mirrors SURVEY.md §1, re-done for MI355X):
  L4  this Python package        (reference: Java API layer)
  L3  pybind11 bindings          (reference: JNI bridge)
  L2  src/gpu/*.hip HIP kernels  (reference: CUDA ops on libcudf)
  L1  memory/OOM state machine   (reference: SparkResourceAdaptorJni.cpp)
  L0  torch caching allocator + HIP runtime (reference: RMM + CUDA)
"""

__version__ = "0.1.0"

from . import columnar  # noqa: F401
from .columnar import Column, DType, Table  # noqa: F401

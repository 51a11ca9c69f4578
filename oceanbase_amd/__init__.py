"""oceanbase_amd — MI355X-native implementation of OceanBase's vectorized OLAP
scan→filter→aggregate hot path (BASELINE.json north_star).

Layout:
  oceanbase_amd.abi     — ctypes mirror of include/obx.h (the C-ABI boundary)
  oceanbase_amd.oracle  — CPU oracle bindings (oracle/liboracle.so).
                          TEST INFRASTRUCTURE + reported CPU baseline only;
                          never the product path.
  oceanbase_amd.engine  — the MI355X product engine bindings (libobx.so,
                          HIP/gfx950). Fails loudly if the HIP library is
                          missing on a GPU machine.
"""

from . import abi  # noqa: F401

__version__ = "0.1.0"

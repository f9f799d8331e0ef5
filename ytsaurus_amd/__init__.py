"""ytsaurus_amd — MI355X-native (gfx950) executor for the YTsaurus
dynamic-table query hot path: columnar scan → filter → hash-aggregate behind
a C-ABI drop-in boundary (include/ytql_gpu.h) mirroring IEvaluator::Run.

Product compute lives in libytql_gpu.so (HIP + C++ host). There is no CPU
fallback: on a machine without a HIP device every execute entry fails with
YT_ERR_NO_GPU. The CPU oracle under oracle/ is test infrastructure only.
"""
from . import _abi        # noqa: F401
from .api import *        # noqa: F401,F403

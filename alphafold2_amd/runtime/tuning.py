"""hipBLASLt solution selection via TunableOp.

A tuning cache for gfx950 (collected with PYTORCH_TUNABLEOP_TUNING=1 on
the flagship training shapes) ships with the package; loading it
read-only picks measured-fastest Tensile solutions (+3.4% whole-step).
Unknown shapes silently use the default heuristics.
"""
import os

import torch

_CACHE = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      'tunableop_gfx950.csv')


def enable_tuned_gemm(cache_path=None):
    """Enable TunableOp with tuning OFF, loading the committed cache.

    No-op on CPU, when the user controls PYTORCH_TUNABLEOP_ENABLED
    themselves, or when the cache is missing.  Returns True if loaded.
    """
    if not torch.cuda.is_available():
        return False
    if os.environ.get('PYTORCH_TUNABLEOP_ENABLED') is not None:
        return False
    path = cache_path or _CACHE
    if not os.path.exists(path):
        return False
    torch.cuda.tunable.enable(True)
    torch.cuda.tunable.tuning_enable(False)
    return torch.cuda.tunable.read_file(path)

"""numpy-safe JSON helpers (parity: /root/reference/maggy/util.py:97)."""
import numpy as np


def json_default_numpy(obj):
    if isinstance(obj, np.integer):
        return int(obj)
    if isinstance(obj, np.floating):
        return float(obj)
    if isinstance(obj, np.bool_):
        return bool(obj)
    if isinstance(obj, np.ndarray):
        return obj.tolist()
    raise TypeError(
        "Object of type {} is not JSON serializable".format(type(obj).__name__)
    )

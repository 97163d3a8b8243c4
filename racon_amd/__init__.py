"""MI355X-native consensus polishing engine (racon capability parity).

The compute engine is native C++/HIP (built in-tree as `_racon` and the
`racon` CLI); this package carries the python-facing utilities: synthetic
data generation for benchmarks, the wrapper script logic, and preprocessing
helpers.
"""

import sys
from pathlib import Path

_REPO = Path(__file__).resolve().parent.parent
for p in (_REPO, _REPO / "build"):
    if str(p) not in sys.path:
        sys.path.insert(0, str(p))


def native():
    """Imports and returns the native _racon module, failing loudly."""
    import _racon

    return _racon

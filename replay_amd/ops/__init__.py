"""HIP-kernel dispatch layer.

Hot ops run through this module: on a ROCm GPU they dispatch to the in-tree
gfx950 HIP extension (``replay_amd/ops/hip``); on CPU they fall back to eager
torch.  On a GPU box a missing extension FAILS LOUDLY instead of silently
running eager (set ``REPLAY_AMD_ALLOW_EAGER=1`` to override for debugging).
"""

from __future__ import annotations

import os
from typing import Optional

_EXT = None
_EXT_TRIED = False


def hip_ext() -> Optional[object]:
    """Load (once) and return the compiled HIP extension, or None on CPU."""
    global _EXT, _EXT_TRIED
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        from replay_amd.ops import _load

        _EXT = _load.load_extension()
    except Exception:  # noqa: BLE001
        _EXT = None
    return _EXT


def require_hip_on_gpu(tensor) -> bool:
    """True -> use the HIP path; False -> eager.  Raises on GPU without ext."""
    if not tensor.is_cuda:
        return False
    ext = hip_ext()
    if ext is None:
        if os.environ.get("REPLAY_AMD_ALLOW_EAGER") == "1":
            return False
        raise RuntimeError(
            "replay_amd HIP extension is not built but a GPU tensor reached a hot op. "
            "Run `python -c 'import __graft_entry__; __graft_entry__.build()'` "
            "or set REPLAY_AMD_ALLOW_EAGER=1 to use the (slow) eager fallback."
        )
    return True

"""Model asset path helpers (parity: reference lib/utils.py:6-10)."""
from __future__ import annotations

import os

from ..config import civitai_cache_dir


def civitai_model_path(model_id: int, version_id: int) -> str:
    """Cache path for a Civitai model file (reference lib/utils.py:6-10)."""
    return os.path.join(civitai_cache_dir(), f"{model_id}_{version_id}.safetensors")

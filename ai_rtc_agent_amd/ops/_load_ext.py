"""Locate and import the in-tree HIP extension (_C*.so next to this file).

Built by `python setup.py build_ext --inplace` (driven by __graft_entry__.build)
with PYTORCH_ROCM_ARCH=gfx950. In-tree (not site-packages) so the .so travels
with the repo snapshot to GPU boxes.
"""
from __future__ import annotations

import glob
import importlib.util
import os


def load():
    here = os.path.dirname(os.path.abspath(__file__))
    cands = sorted(glob.glob(os.path.join(here, "_C*.so")))
    if not cands:
        raise ImportError(f"no built _C*.so under {here}")
    spec = importlib.util.spec_from_file_location("ai_rtc_agent_amd.ops._C", cands[0])
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod

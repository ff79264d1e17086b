"""
Scan engines: the pluggable executors of the record pipeline.

  * cpu: the pure-Python oracle (always available; the semantic reference)
  * gpu: the MI355X HIP engine (fused CDNA4 kernel; requires the compiled
         extension and a visible GPU)

Selection: DRAGNET_ENGINE=cpu|gpu overrides; default is gpu when torch
reports a GPU, else cpu.  On a GPU machine the GPU engine refuses to fall
back silently — a missing extension is a hard error.
"""

import os


def get_engine(name=None):
    name = name or os.environ.get("DRAGNET_ENGINE")
    if name is None:
        name = "gpu" if _gpu_available() else "cpu"
    if name == "cpu":
        from .cpu import CpuEngine
        return CpuEngine()
    if name == "gpu":
        from .gpu import GpuEngine
        return GpuEngine()
    raise ValueError('unknown engine: "%s"' % name)


def _gpu_available():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False

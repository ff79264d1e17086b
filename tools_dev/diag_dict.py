"""Dict sizing sweep on the high-card urls shape (committed to
tools_dev if it shows anything)."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from dragnet_amd.engine import plan as planmod
from dragnet_amd.engine.gpu import GpuEngine, _ScanContext
from dragnet_amd.query import query_load
from dragnet_amd.tools.mktestdata import generate_lines

lines = []
total = 0
for line in generate_lines(1 << 62, seed=9):
    lines.append(line)
    total += len(line)
    if total >= 256 << 20:
        break
pool = b"".join(lines)
nrec = len(lines)
eng = GpuEngine()
eng.chunk_bytes = len(pool)
CASES = [("urls", query_load(breakdown_specs="req.url")),
         ("flagship", query_load(
             filter={"eq": ["req.method", "GET"]},
             breakdown_specs="req.method,res.statusCode"))]
for dslots in (18, 20, 22):
    for name, q in CASES:
        cplan = planmod.compile_plan([q])
        ctx = _ScanContext(eng, cplan, 1 << 18, 1 << dslots, 64 << 20)
        ctx.stage_xpose(pool)
        def one(c=ctx):
            c.reset()
            c.scan_xpose()
        for _ in range(2):
            one()
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(6):
            one()
        torch.cuda.synchronize()
        dt = (time.time() - t0) / 6
        print("dict=1<<%d %-9s %7.1f GB/s %8.1f M rec/s"
              % (dslots, name, len(pool)/dt/1e9, nrec/dt/1e6), flush=True)
        del ctx

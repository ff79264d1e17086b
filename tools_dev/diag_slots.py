"""Dense-directory slot sweep on the multi-thousand-group shapes:
probe-chain length vs directory size (DRAGNET_DENSE_SLOTS)."""
import os
import sys
import time

sys.path.insert(0, "/root/repo")

import torch  # noqa: E402

from dragnet_amd.engine import plan as planmod  # noqa: E402
from dragnet_amd.engine.gpu import GpuEngine, _ScanContext  # noqa: E402
from dragnet_amd.query import query_load  # noqa: E402
from dragnet_amd.tools.mktestdata import generate_lines  # noqa: E402

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

CASES = [
    ("5-field bd", query_load(
        breakdown_specs="host,operation,req.method,res.statusCode,"
                        "latency[aggr=quantize]")),
    ("flagship filter+2bd", query_load(
        filter={"eq": ["req.method", "GET"]},
        breakdown_specs="req.method,res.statusCode")),
]

for slots in (8192, 16384, 32768):
    os.environ["DRAGNET_DENSE_SLOTS"] = str(slots)
    for mode in ("linear", "xpose"):
        for name, q in CASES:
            cplan = planmod.compile_plan([q])
            ctx = _ScanContext(eng, cplan, 1 << 18, 1 << 18, 32 << 20)
            if mode == "xpose":
                ctx.stage_xpose(pool)

                def one(c=ctx):
                    c.reset()
                    c.scan_xpose()
            else:
                ctx.stage_resident(pool)

                def one(c=ctx):
                    c.reset()
                    c.scan_resident(h2d=False)
            for _ in range(2):
                one()
            torch.cuda.synchronize()
            t0 = time.time()
            for _ in range(6):
                one()
            torch.cuda.synchronize()
            dt = (time.time() - t0) / 6
            print("slots=%-6d %-7s %-22s %7.1f GB/s %8.1f M rec/s"
                  % (slots, mode, name, len(pool) / dt / 1e9,
                     nrec / dt / 1e6), flush=True)
            del ctx

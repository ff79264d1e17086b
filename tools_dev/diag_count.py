import sys, time
sys.path.insert(0, "/root/repo")
import torch
from dragnet_amd.engine import plan as planmod
from dragnet_amd.engine.gpu import GpuEngine, _ScanContext
from dragnet_amd.query import query_load
from dragnet_amd.tools.mktestdata import generate_lines
lines = []; total = 0
for line in generate_lines(1 << 62, seed=9):
    lines.append(line); total += len(line)
    if total >= 256 << 20: break
pool = b"".join(lines); nrec = len(lines)
eng = GpuEngine(); eng.chunk_bytes = len(pool)
for name, q in [("count-only", query_load()),
                ("count w/ 1 field cap", query_load(filter={"eq":["req.method","GET"]})),
                ("flagship", query_load(filter={"eq":["req.method","GET"]}, breakdown_specs="req.method,res.statusCode"))]:
    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(eng, cplan, 1 << 16, 1 << 16, 32 << 20)
    ctx.stage_resident(pool)
    for _ in range(2): ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(6): ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize(); dt = (time.time()-t0)/6
    print("%-24s %7.2f GB/s" % (name, len(pool)/dt/1e9))

# date-parse heavy shape (reference headline histogram case);
# note: pool timestamps are ~constant (generator quirk) but every
# record still runs the full ISO parse
for name, q in [("daily date histogram",
                 query_load(breakdown_specs="ts[date,field=time,aggr=lquantize,step=86400]")),
                ("date+2 fields",
                 query_load(breakdown_specs="ts[date,field=time,aggr=lquantize,step=3600],req.method,res.statusCode"))]:
    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(eng, cplan, 1 << 16, 1 << 16, 32 << 20)
    ctx.stage_resident(pool)
    for _ in range(2): ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(6): ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize(); dt = (time.time()-t0)/6
    print("%-24s %7.2f GB/s" % (name, len(pool)/dt/1e9))

# Diagnostic: which query shapes bound the resident scan kernel?
import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from dragnet_amd.engine import plan as planmod
from dragnet_amd.engine.gpu import GpuEngine, _ScanContext
from dragnet_amd.query import query_load
from dragnet_amd.tools.mktestdata import generate_lines

lines = []
total = 0
for line in generate_lines(1 << 62, seed=9):
    lines.append(line); total += len(line)
    if total >= 256 << 20: break
pool = b"".join(lines)
nrec = len(lines)
eng = GpuEngine()
eng.chunk_bytes = len(pool)

CASES = [
  ("count-only (no fields)", query_load()),
  ("flagship: filter + 2 breakdowns", query_load(filter={"eq":["req.method","GET"]}, breakdown_specs="req.method,res.statusCode")),
  ("pure-ordinal breakdown (no interns)", query_load(breakdown_specs="dataLatency[aggr=lquantize,step=100]")),
  ("1 string breakdown", query_load(breakdown_specs="req.method")),
  ("high-card string (urls)", query_load(breakdown_specs="req.url")),
  ("5-field breakdown", query_load(breakdown_specs="host,operation,req.method,res.statusCode,latency[aggr=quantize]")),
  ("filter-only count", query_load(filter={"eq":["req.method","GET"]})),
]
for name, q in CASES:
    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(eng, cplan, 1 << 18, 1 << 18, 32 << 20)
    ctx.stage_resident(pool)
    for _ in range(2):
        ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(6):
        ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 6
    print("%-38s %7.2f GB/s  %6.1f M rec/s" % (name, len(pool)/dt/1e9, nrec/dt/1e6))

# ablation for the slow ordinal case
CASES2 = [
  ("dataLatency lquantize100", query_load(breakdown_specs="dataLatency[aggr=lquantize,step=100]")),
  ("dataLatency quantize(p2)", query_load(breakdown_specs="dataLatency[aggr=quantize]")),
  ("dataLatency plain (num intern)", query_load(breakdown_specs="dataLatency")),
  ("latency(str) plain", query_load(breakdown_specs="latency")),
  ("res.statusCode plain", query_load(breakdown_specs="res.statusCode")),
]
for name, q in CASES2:
    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(eng, cplan, 1 << 18, 1 << 18, 32 << 20)
    ctx.stage_resident(pool)
    for _ in range(2):
        ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(6):
        ctx.reset(); ctx.scan_resident(h2d=False)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 6
    print("%-38s %7.2f GB/s  %6.1f M rec/s" % (name, len(pool)/dt/1e9, nrec/dt/1e6))

# raw H2D link rate + finalize cost
pin = torch.empty(256 << 20, dtype=torch.uint8, pin_memory=True)
dev = torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
for _ in range(3):
    dev.copy_(pin, non_blocking=True)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(10):
    dev.copy_(pin, non_blocking=True)
torch.cuda.synchronize()
print("raw H2D: %.2f GB/s" % ((256 << 20) * 10 / (time.time() - t0) / 1e9))

q = CASES[1][1]
cplan = planmod.compile_plan([q])
ctx = _ScanContext(eng, cplan, 1 << 18, 1 << 18, 32 << 20)
ctx.stage_resident(pool)
ctx.reset(); ctx.scan_resident(h2d=False); ctx.finalize([q])
t0 = time.time()
for _ in range(10):
    ctx.finalize([q])
print("finalize: %.2f ms" % ((time.time() - t0) / 10 * 1000))

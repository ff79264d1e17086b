"""Extended fuzz soak: more seeds than the CI suite (run on demand)."""
import sys
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")
import os, random, tempfile
from test_gpu_fuzz import rand_line, rand_query
from dragnet_amd.engine.cpu import CpuEngine
from dragnet_amd.engine.gpu import GpuEngine

cpu, gpu = CpuEngine(), GpuEngine()
fails = 0
N = int(os.environ.get("SOAK_SEEDS", "32"))
for seed in range(100, 100 + N):
    rng = random.Random(seed)
    lines = [rand_line(rng) for _ in range(1500)]
    with tempfile.NamedTemporaryFile(suffix=".ndjson", delete=False) as f:
        f.write(b"\n".join(lines) + b"\n")
        path = f.name
    try:
        for qi in range(5):
            q = rand_query(rng)
            c = cpu.scan([path], [q])
            g = gpu.scan([path], [q])
            if g.aggregators[0].points() != c.aggregators[0].points():
                fails += 1
                print("DIVERGENCE seed=%d qi=%d filter=%r bds=%r"
                      % (seed, qi, q.filter,
                         [b["name"] for b in q.breakdowns]))
    finally:
        os.unlink(path)
print("soak done: %d divergences over %d seeds x 5 queries" % (fails, N))
assert fails == 0

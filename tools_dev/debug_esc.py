import sys
sys.path.insert(0, "/root/repo")
from dragnet_amd.engine.cpu import CpuEngine
from dragnet_amd.engine.gpu import GpuEngine
from dragnet_amd.query import query_load
data = (b'{"m": "a\\nb", "x": 1}\n'
        b'{"m": "a\\u000ab", "x": 2}\n'
        b'{"m": "anb", "x": 3}\n'
        b'{"m": "quote\\"q", "x": 4}\n')
open("/tmp/esc.ndjson", "wb").write(data)
print("data:", data)
cpu, gpu = CpuEngine(), GpuEngine()
for filt in [{"eq": ["m", "a\nb"]}, {"eq": ["m", 'quote"q']}, None]:
    q = query_load(filter=filt, breakdown_specs="m" if filt is None else None)
    c = cpu.scan(["/tmp/esc.ndjson"], [q])
    g = gpu.scan(["/tmp/esc.ndjson"], [q])
    print("filt", filt)
    print("  cpu:", c.aggregators[0].points())
    print("  gpu:", g.aggregators[0].points())
    print("  gpu stages:", g.stages)

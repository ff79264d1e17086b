"""Diagnose the envelope-fuzz ninputs overcount: compare the device
newline index against torch's own nonzero() over the same bytes, and
trace where the extra record comes from."""
import json
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def fixture():
    rng = random.Random(99)
    lines = []
    for i in range(4000):
        r = rng.random()
        if r < 0.2:
            d = rng.randint(60, 90)
            lines.append((("{\"k\":" * d) + '"v"' + ("}" * d)).encode())
        elif r < 0.4:
            digits = "".join(rng.choice("123456789")
                             for _ in range(rng.randint(18, 30)))
            lines.append(('{"n": %s}' % digits).encode())
        elif r < 0.55:
            lines.append(b'{"\\u006b%d": "x"}' % (i % 7))
        elif r < 0.7:
            lines.append(bytes(rng.randrange(256)
                               for _ in range(rng.randint(1, 40)))
                         .replace(b"\n", b"x"))
        else:
            lines.append(json.dumps(
                {"k%d" % (i % 7): rng.choice(["a", "b", 3, None]),
                 "n": rng.randint(0, 99)}).encode())
    return b"".join(ln + b"\n" for ln in lines)


def main():
    import torch

    from dragnet_amd.engine.gpu import GpuEngine, _ScanContext, _pad
    from dragnet_amd.engine import plan as planmod
    from dragnet_amd.query import query_load

    data = fixture()
    n = len(data)
    eng = GpuEngine()
    q = query_load(breakdown_specs="n")
    cplan = planmod.compile_plan([q])
    ctx = _ScanContext(eng, cplan, 1 << 16, 1 << 16, 32 << 20)
    ctx._ensure_buffers(_pad(n))
    pin = ctx._pinned
    pin[:n] = torch.frombuffer(bytearray(data), dtype=torch.uint8)
    padded = _pad(n)
    pin[n:padded] = 10
    dev = ctx._dev_data
    dev[:padded].copy_(pin[:padded])
    torch.cuda.synchronize()

    for rep in range(5):
        eng.ops.newline_index(dev, 0, n, ctx._segs, ctx._pos,
                              ctx._nlines)
        torch.cuda.synchronize()
        nl = int(ctx._nlines.item())
        truth = (dev[:n] == 10).nonzero().flatten()
        print("rep %d: device count=%d torch count=%d" %
              (rep, nl, truth.numel()))
        if nl != truth.numel():
            pos = ctx._pos[:nl].to(torch.int64)
            t = truth
            m = min(nl, t.numel())
            diff = (pos[:m] != t[:m]).nonzero().flatten()
            i = int(diff[0].item()) if diff.numel() else m
            print("  first divergence at index", i)
            print("  device pos[%d-5:%d+5] =" % (i, i),
                  pos[max(0, i - 5):i + 5].tolist())
            print("  torch  pos[%d-5:%d+5] =" % (i, i),
                  t[max(0, i - 5):i + 5].tolist())
            # dup or phantom?
            import numpy as np
            pn = pos.cpu().numpy()
            dup = pn[1:][pn[1:] == pn[:-1]]
            print("  duplicate positions:", dup[:10])
            sd = np.setdiff1d(pn, t.cpu().numpy())
            print("  phantom positions:", sd[:10],
                  [bytes(data[max(0, int(x) - 8):int(x) + 8])
                   for x in sd[:3]])

    # full engine scan counters
    import tempfile
    with tempfile.NamedTemporaryFile(suffix=".log", delete=False) as f:
        f.write(data)
        path = f.name
    g = eng.scan([path], [q])
    print("engine stages:", dict(g.stages)["json parser"])


if __name__ == "__main__":
    main()

"""
BASELINE configs #2-#4 end-to-end measurement on one MI355X
(VERDICT r1 "Next round" #2):

  #2  dn scan -b req.method over ~10 GB synthetic NDJSON
  #3  dn scan -f <filter> -b req.method,res.statusCode over the
      largest synthetic corpus that safely fits (target 100 GB-class)
  #4  dn build + dn query with examples/index-muskie-local.json

Data: mktestdata-shaped records; 2 GB of distinct generated data
replicated file-wise to the target size (every byte is still read,
parsed and aggregated — duplication does not change scan work).
Corpus lives in /dev/shm when it fits with a wide safety margin,
else under ./gpurun_out-adjacent scratch.  Results printed as JSON
lines and written to gpurun_out/configs_r02.json.
"""
import json
import os
import shutil
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

GB = 1 << 30


def log(msg):
    sys.stderr.write("[measure] %s\n" % msg)
    sys.stderr.flush()


def gen_distinct(dest, total_bytes, nfiles=2):
    """Generate `nfiles` distinct NDJSON files totalling total_bytes."""
    from dragnet_amd.tools.mktestdata import generate_lines
    per = total_bytes // nfiles
    files = []
    nrec = 0
    for i in range(nfiles):
        path = os.path.join(dest, "gen_%02d.log" % i)
        t0 = time.time()
        with open(path, "wb") as f:
            size = 0
            buf = []
            for line in generate_lines(1 << 62, seed=4242 + i):
                buf.append(line)
                size += len(line)
                nrec += 1
                if len(buf) >= 20000:
                    f.write(b"".join(buf))
                    buf = []
                if size >= per:
                    break
            f.write(b"".join(buf))
        log("generated %s: %.2f GB in %.0fs"
            % (path, size / GB, time.time() - t0))
        files.append((path, size))
    return files, nrec


def replicate(files, dest, target_bytes):
    """File-level replication up to target_bytes; returns total bytes
    and the record multiplier."""
    base = sum(sz for _p, sz in files)
    total = base
    copies = 1
    i = 0
    t0 = time.time()
    while total + base <= target_bytes:
        for p, sz in files:
            dst = os.path.join(dest, "rep_%03d_%s"
                               % (i, os.path.basename(p)))
            # reflink-less copy; tmpfs = memcpy speed
            shutil.copyfile(p, dst)
            total += sz
        copies += 1
        i += 1
    log("corpus: %.2f GB (%dx) in %.0fs"
        % (total / GB, copies, time.time() - t0))
    return total, copies


def run_dn(args, env=None, timing=False):
    cmd = [sys.executable, "-m", "dragnet_amd.cli"]
    if timing:
        cmd.append("-t")
    cmd += args
    t0 = time.time()
    r = subprocess.run(cmd, capture_output=True, text=True, env=env)
    wall = time.time() - t0
    if r.returncode != 0:
        log("dn %s FAILED: %s" % (args[:2], r.stderr[-2000:]))
        raise SystemExit(1)
    return wall, r.stdout, r.stderr


def api_scan_time(data_dir, filter_=None, breakdowns="req.method",
                  readers=24):
    """In-process engine-level scan timing (excludes interpreter/torch
    startup, which the CLI wall numbers include)."""
    os.environ["DRAGNET_READERS"] = str(readers)
    from dragnet_amd.config import Datasource
    from dragnet_amd.datasource.file import FileDatasource
    from dragnet_amd.engine import get_engine
    from dragnet_amd.query import query_load
    ds = Datasource(name="t", backend="file", path=data_dir,
                    time_field="time")
    fd = FileDatasource(ds, engine=get_engine())
    q = query_load(filter=filter_, breakdown_specs=breakdowns)
    t0 = time.time()
    r = fd.scan(q)
    dt = time.time() - t0
    n = dict(r.stages)["json parser"]["ninputs"]
    return dt, n


def main():
    out_path = os.path.join(REPO, "gpurun_out", "configs_r02.json")
    results = []

    # ---- placement ----
    shm = shutil.disk_usage("/dev/shm")
    log("/dev/shm: %.0f GB free of %.0f GB"
        % (shm.free / GB, shm.total / GB))
    # wide margins: never take more than half the free space, never
    # more than free-64GB
    budget = min(shm.free // 2, shm.free - 64 * GB)
    root = "/dev/shm/dn_corpus"
    if budget < 12 * GB:
        root = os.path.join(REPO, "gpurun_out", "dn_corpus")
        du = shutil.disk_usage(os.path.dirname(root))
        budget = min(du.free // 2, du.free - 32 * GB)
        log("using disk scratch, budget %.0f GB" % (budget / GB))
    os.makedirs(root, exist_ok=True)

    data10 = os.path.join(root, "d10")
    os.makedirs(data10, exist_ok=True)
    base_b = int(float(os.environ.get("MC_BASE_GB", 2)) * GB)
    t10_b = int(float(os.environ.get("MC_T10_GB", 10)) * GB)
    files, nrec_base = gen_distinct(data10, base_b)
    bytes10, copies10 = replicate(files, data10, t10_b)
    nrec10 = nrec_base * copies10

    cfg = os.path.join(root, "dnrc.json")
    env = dict(os.environ)
    env["DRAGNET_CONFIG"] = cfg
    idx_root = os.path.join(root, "idx")

    run_dn(["datasource-add", "d10", "--path=" + data10,
            "--time-field=time", "--index-path=" + idx_root], env=env)

    # ---- config #2: dn scan -b req.method, 10 GB ----
    for readers in (8, 16, 24, 32):
        env["DRAGNET_READERS"] = str(readers)
        wall, out, err = run_dn(
            ["scan", "-b", "req.method", "d10"], env=env, timing=True)
        # subtract interpreter+import startup (the 'require' phase)
        req = 0.0
        for ln in err.splitlines():
            if "require:" in ln:
                req = float(ln.split()[-1].rstrip("s"))
        res = {
            "config": "#2 dn scan -b req.method",
            "bytes": bytes10, "records": nrec10,
            "wall_s": round(wall, 3),
            "require_s": round(req, 3),
            "readers": readers,
            "gb_per_sec": round(bytes10 / (wall - req) / 1e9, 2),
            "recs_per_sec": round(nrec10 / (wall - req), 0),
        }
        log(json.dumps(res))
        results.append(res)

    # engine-level timing for #2 (no startup), reader sweep
    for readers in (8, 16, 24, 32):
        dt, nrec = api_scan_time(data10, readers=readers)
        res = {"config": "#2 engine-level scan", "readers": readers,
               "bytes": bytes10, "records": nrec,
               "scan_s": round(dt, 3),
               "gb_per_sec": round(bytes10 / dt / 1e9, 2),
               "recs_per_sec": round(nrec / dt, 0)}
        log(json.dumps(res))
        results.append(res)

    # ---- config #3: filter + 2-field breakdown, as large as fits ---
    target3 = min(budget - bytes10,
                  int(float(os.environ.get("MC_T100_GB", 100)) * GB))
    data100 = os.path.join(root, "d100")
    os.makedirs(data100, exist_ok=True)
    for p, _sz in files:
        shutil.copyfile(p, os.path.join(data100,
                                        os.path.basename(p)))
    bytes100, copies100 = replicate(files, data100, target3)
    nrec100 = nrec_base * copies100
    run_dn(["datasource-add", "d100", "--path=" + data100,
            "--time-field=time"], env=env)
    env["DRAGNET_READERS"] = os.environ.get("MC_READERS", "24")
    wall, out, err = run_dn(
        ["scan", "-f", '{"eq": ["req.method", "GET"]}',
         "-b", "req.method,res.statusCode", "d100"],
        env=env, timing=True)
    req = 0.0
    for ln in err.splitlines():
        if "require:" in ln:
            req = float(ln.split()[-1].rstrip("s"))
    res = {
        "config": "#3 dn scan filter + 2-field breakdown",
        "bytes": bytes100, "records": nrec100,
        "wall_s": round(wall, 3), "require_s": round(req, 3),
        "gb_per_sec": round(bytes100 / (wall - req) / 1e9, 2),
        "recs_per_sec": round(nrec100 / (wall - req), 0),
    }
    log(json.dumps(res))
    results.append(res)
    dt, nrec = api_scan_time(
        data100, filter_={"eq": ["req.method", "GET"]},
        breakdowns="req.method,res.statusCode",
        readers=int(os.environ.get("MC_READERS", "24")))
    res = {"config": "#3 engine-level scan",
           "bytes": bytes100, "records": nrec,
           "scan_s": round(dt, 3),
           "gb_per_sec": round(bytes100 / dt / 1e9, 2),
           "recs_per_sec": round(nrec / dt, 0)}
    log(json.dumps(res))
    results.append(res)

    # ---- config #5 shape (single-node leg): sharded-backend scan
    # over the same corpus; world=1 here — the 8-rank DP + RCCL merge
    # leg is the driver's SCALE run (bench.py is exactly that shape)
    run_dn(["datasource-add", "d100s", "--backend=sharded",
            "--path=" + data100, "--time-field=time"], env=env)
    wall, out, err = run_dn(
        ["scan", "-f", '{"eq": ["req.method", "GET"]}',
         "-b", "req.method,res.statusCode", "d100s"],
        env=env, timing=True)
    req = 0.0
    for ln in err.splitlines():
        if "require:" in ln:
            req = float(ln.split()[-1].rstrip("s"))
    res = {
        "config": "#5 sharded-backend scan (world=1 leg)",
        "bytes": bytes100, "records": nrec100,
        "wall_s": round(wall, 3), "require_s": round(req, 3),
        "gb_per_sec": round(bytes100 / (wall - req) / 1e9, 2),
    }
    log(json.dumps(res))
    results.append(res)

    # ---- config #4: build + query with the muskie index ----
    wall_b, out, err = run_dn(
        ["build", "--index-config",
         os.path.join(REPO, "examples", "index-muskie-local.json"),
         "d10"], env=env, timing=True)
    req = 0.0
    for ln in err.splitlines():
        if "require:" in ln:
            req = float(ln.split()[-1].rstrip("s"))
    wall_q, out_q, _ = run_dn(
        ["query", "-b", "req.method,res.statusCode",
         "-f", '{"eq": ["req.method", "GET"]}', "d10"], env=env)
    wall_s, out_s, _ = run_dn(
        ["scan", "-b", "req.method,res.statusCode",
         "-f", '{"eq": ["req.method", "GET"]}', "d10"], env=env)
    res = {
        "config": "#4 dn build + dn query (muskie index), 1 GPU",
        "bytes": bytes10, "records": nrec10,
        "build_wall_s": round(wall_b, 3),
        "build_require_s": round(req, 3),
        "build_gb_per_sec": round(bytes10 / (wall_b - req) / 1e9, 2),
        "query_wall_s": round(wall_q, 3),
        "query_equals_scan": out_q == out_s,
    }
    log(json.dumps(res))
    results.append(res)

    with open(out_path, "w") as f:
        json.dump(results, f, indent=2)
    log("wrote %s" % out_path)
    shutil.rmtree(root, ignore_errors=True)


if __name__ == "__main__":
    main()

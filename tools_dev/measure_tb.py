"""TB-class single-rank file-scan leg (BASELINE config #5's scale
label, single-GPU slice): build the largest corpus /dev/shm safely
holds (capped at 1 TB), scan it engine-level with the flagship query,
report GB/s.

DO NOT run this on a shared or RAM-constrained box.  The first
attempt used tmpfs free space as the safety margin and took the box
down: tmpfs `disk_usage().free` reflects the MOUNT size, not actual
available RAM, so filling it can OOM the host.  The guard below now
keys off /proc/meminfo MemAvailable with a 256 GB floor and a 60%
cap, and the corpus is removed afterwards — but treat this tool as
operator-supervised, not CI."""
import json
import multiprocessing as mp
import os
import shutil
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from measure_configs import (GB, api_scan_time, gen_distinct,  # noqa: E402
                             log)

ROOT = "/dev/shm/dn_tb"


def _copy_batch(jobs):
    for src, dst in jobs:
        shutil.copyfile(src, dst)
    return len(jobs)


def mem_available_bytes():
    with open("/proc/meminfo") as f:
        for ln in f:
            if ln.startswith("MemAvailable:"):
                return int(ln.split()[1]) * 1024
    return 0


def main():
    shm = shutil.disk_usage("/dev/shm")
    avail = mem_available_bytes()
    log("/dev/shm free: %.0f GB of %.0f; MemAvailable: %.0f GB"
        % (shm.free / GB, shm.total / GB, avail / GB))
    # tmpfs free space is NOT available RAM — key the budget off
    # MemAvailable (what the kernel can actually give us)
    target = min(shm.free - 192 * GB,
                 int(avail * 0.6),
                 avail - 256 * GB,
                 int(float(os.environ.get("TB_TARGET_GB", 1024)) * GB))
    if target < 300 * GB:
        log("not enough ACTUAL free RAM for a TB-class corpus; "
            "refusing (need >= ~560 GB MemAvailable)")
        return 1
    shutil.rmtree(ROOT, ignore_errors=True)
    os.makedirs(ROOT)
    try:
        files, nrec_base = gen_distinct(ROOT, 2 * GB, nfiles=4)
        base = sum(sz for _p, sz in files)
        ncopies = max(0, int(target // base) - 1)
        jobs = []
        for i in range(ncopies):
            for p, _sz in files:
                jobs.append((p, os.path.join(
                    ROOT, "rep_%04d_%s" % (i, os.path.basename(p)))))
        t0 = time.time()
        nproc = 16
        slices = [jobs[k::nproc] for k in range(nproc)]
        with mp.Pool(nproc) as pool:
            pool.map(_copy_batch, slices)
        total = base * (ncopies + 1)
        log("corpus: %.1f GB (%d files) replicated in %.0fs"
            % (total / GB, len(jobs) + len(files), time.time() - t0))

        readers = int(os.environ.get("TB_READERS", "24"))
        dt, nrec = api_scan_time(
            ROOT, filter_={"eq": ["req.method", "GET"]},
            breakdowns="req.method,res.statusCode", readers=readers)
        res = {
            "config": "#5 single-rank TB-class leg "
                      "(engine-level file scan, flagship query)",
            "bytes": total, "records": nrec, "scan_s": round(dt, 3),
            "gb_per_sec": round(total / dt / 1e9, 2),
            "recs_per_sec": round(nrec / dt, 0),
            "readers": readers,
        }
        assert nrec == nrec_base * (ncopies + 1), (nrec, nrec_base)
        log(json.dumps(res))
        out = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "gpurun_out", "tb_r02.json")
        os.makedirs(os.path.dirname(out), exist_ok=True)
        with open(out, "w") as f:
            f.write(json.dumps(res) + "\n")
        return 0
    finally:
        shutil.rmtree(ROOT, ignore_errors=True)


if __name__ == "__main__":
    sys.exit(main())

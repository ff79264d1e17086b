"""TB-class single-rank file-scan leg (BASELINE config #5's scale
label, single-GPU slice): build the largest corpus /dev/shm safely
holds (capped at 1 TB), scan it engine-level with the flagship query,
report GB/s.  Refuses to run without a ~192 GB free-RAM margin and
removes the corpus afterwards."""
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


def main():
    shm = shutil.disk_usage("/dev/shm")
    log("/dev/shm free: %.0f GB of %.0f" % (shm.free / GB,
                                            shm.total / GB))
    target = min(shm.free - 192 * GB,
                 int(float(os.environ.get("TB_TARGET_GB", 1024)) * GB))
    if target < 300 * GB:
        log("not enough /dev/shm headroom for a TB-class corpus; "
            "refusing (need >= ~500 GB free)")
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

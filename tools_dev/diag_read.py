"""Host read-path microbench: parallel preadv from /dev/shm into a
pinned buffer at various reader counts — isolates the fill bandwidth
that bounds `dn scan` over real files."""
import concurrent.futures as cf
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def main():
    import torch
    size = 4 << 30
    path = "/dev/shm/readbench.bin"
    with open(path, "wb") as f:
        chunk = os.urandom(1 << 20) * 16  # 16 MB pattern
        for _ in range(size // len(chunk)):
            f.write(chunk)
    pin = torch.empty(256 << 20, dtype=torch.uint8, pin_memory=True)
    mv = memoryview(pin.numpy())
    fd = os.open(path, os.O_RDONLY)

    def pread_full(mv2, off):
        done = 0
        while done < len(mv2):
            got = os.preadv(fd, [mv2[done:]], off + done)
            if got <= 0:
                break
            done += got
        return done

    for nr in (1, 4, 8, 16, 32, 48):
        pool = cf.ThreadPoolExecutor(max_workers=nr)
        t0 = time.time()
        total = 0
        off = 0
        while off < size:
            want = min(256 << 20, size - off)
            sec = (want + nr - 1) // nr
            futs = []
            for s in range(0, want, sec):
                e = min(s + sec, want)
                futs.append(pool.submit(pread_full, mv[s:e], off + s))
            for f in futs:
                total += f.result()
            off += want
        dt = time.time() - t0
        print("readers=%2d: %.1f GB/s" % (nr, total / dt / 1e9))
        pool.shutdown()
    os.close(fd)
    os.unlink(path)


if __name__ == "__main__":
    main()

"""
dn-launch: build and run the distributed job for a `dn` command — the
in-node analog of the reference's Manta job builder + CLI-args
serialization (reference lib/manta-job-builder.js:10-55,
lib/datasource-manta.js:774-896): where the reference packages a query
back into shell-escaped `dn` CLI arguments inside a Manta job
definition, this builds the torch.distributed.run invocation that runs
one `dn` rank per GPU over RCCL.

    python -m dragnet_amd.launch [--gpus N] [--dry-run] SUBCOMMAND ...

--dry-run prints the job definition as JSON (the reference's dry-run
prints its Manta job JSON the same way, tests/dn/manta/tst.scan_manta.sh).
"""

import json
import shlex
import subprocess
import sys


def sh_escape(arg):
    """Shell-quote one argument (reference shEscape,
    lib/datasource-manta.js:887-896)."""
    return shlex.quote(arg)


def build_job(dn_args, nproc, master_port=0):
    """Build the job definition for running `dn <dn_args>` as nproc
    ranks (one per GPU) on this node."""
    if master_port == 0:
        import socket
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        master_port = s.getsockname()[1]
        s.close()
    phase_cmd = [sys.executable, "-m", "dragnet_amd.cli"] + dn_args
    launcher = [sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", "--nproc-per-node", str(nproc),
                "--master-addr", "127.0.0.1",
                "--master-port", str(master_port),
                "--no-python", "--"] + phase_cmd
    return {
        "name": "dragnet %s" % (dn_args[0] if dn_args else ""),
        "nprocPerNode": nproc,
        "phases": [{
            "type": "rank",
            "exec": " ".join(sh_escape(a) for a in phase_cmd),
        }],
        "cmd": launcher,
    }


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    nproc = 1
    dry_run = False
    while argv and argv[0].startswith("--"):
        if argv[0] == "--gpus":
            argv.pop(0)
            nproc = int(argv.pop(0))
        elif argv[0].startswith("--gpus="):
            nproc = int(argv.pop(0).split("=", 1)[1])
        elif argv[0] == "--dry-run":
            argv.pop(0)
            dry_run = True
        elif argv[0] == "--help":
            sys.stdout.write(
                "usage: python -m dragnet_amd.launch [--gpus N] "
                "[--dry-run] SUBCOMMAND ...\n"
                "Runs `dn SUBCOMMAND ...` as N ranks (one per GPU) "
                "via torch.distributed.run;\n--dry-run prints the "
                "job definition as JSON.\n")
            return 0
        else:
            sys.stderr.write("dn-launch: unknown option %s\n" % argv[0])
            return 2
    if not argv:
        sys.stderr.write(
            "usage: python -m dragnet_amd.launch [--gpus N] "
            "[--dry-run] SUBCOMMAND ...\n")
        return 2
    job = build_job(argv, nproc)
    if dry_run:
        shown = {k: v for k, v in job.items() if k != "cmd"}
        print(json.dumps(shown, indent=4))
        return 0
    if nproc == 1:
        return subprocess.call(
            [sys.executable, "-m", "dragnet_amd.cli"] + argv)
    return subprocess.call(job["cmd"])


if __name__ == "__main__":
    sys.exit(main())

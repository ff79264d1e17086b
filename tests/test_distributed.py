"""
Multi-process distributed tests (gloo backend, CPU, world_size 2):
sharded scan + merge must equal a single-process scan of all files —
the DP/RCCL path made correct by construction (associative point
merge), exercised here without a GPU.
"""

import os
import sys

import pytest
import torch.multiprocessing as mp


def _worker_scan(rank, world, port, files, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DRAGNET_ENGINE"] = "cpu"
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))

    from dragnet_amd.distributed import (init_process_group,
                                         merge_counter_stages,
                                         merge_points_object,
                                         merge_tables_tensor,
                                         shard_files)
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load

    init_process_group(backend="gloo")
    q = query_load(filter={"eq": ["req.method", "GET"]},
                   breakdown_specs="operation,res.statusCode")
    mine = shard_files(files, rank, world)
    res = CpuEngine().scan(mine, [q])

    merged = merge_points_object(res.aggregators, [q])
    stages = merge_counter_stages(res.stages)

    # tensor-path merge gives every rank the full result
    import torch
    full = merge_tables_tensor(res.aggregators[0], q,
                               torch.device("cpu"))

    if rank == 0:
        out_q.put({
            "points": merged[0].points(),
            "stages": stages,
            "tensor_points": full.points(),
        })
    else:
        out_q.put({"tensor_points": full.points()})

    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_sharded_scan_merge(fixture_tree, tmp_path):
    files = []
    for root, _dirs, names in os.walk(fixture_tree):
        for n in sorted(names):
            files.append(os.path.join(root, n))
    files.sort()

    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load
    q = query_load(filter={"eq": ["req.method", "GET"]},
                   breakdown_specs="operation,res.statusCode")
    single = CpuEngine().scan(files, [q])
    expected = single.aggregators[0].points()

    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    world = 2
    port = 29531
    procs = [ctx.Process(target=_worker_scan,
                         args=(r, world, port, files, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [out_q.get(timeout=90) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    root_res = next(r for r in results if "points" in r)
    assert root_res["points"] == expected
    # tensor allgather path: every rank got the full result
    for r in results:
        assert r["tensor_points"] == expected

    # merged parser counters equal the single-process scan's
    single_parser = dict(single.stages)["json parser"]
    merged_parser = dict(root_res["stages"])["json parser"]
    assert merged_parser == single_parser


def _worker_cli(rank, world, port, cfgfile, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DRAGNET_CONFIG"] = cfgfile
    os.environ["DRAGNET_ENGINE"] = "cpu"
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import io

    from dragnet_amd import cli
    buf = io.StringIO()
    old = sys.stdout
    sys.stdout = buf
    try:
        code = cli.main(["scan", "-b", "req.method", "shard_src"])
    finally:
        sys.stdout = old
    out_q.put((rank, code, buf.getvalue()))
    import torch.distributed as dist
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_sharded_cli(fixture_tree, tmp_path):
    """`dn scan` on a sharded datasource under a 2-rank world: rank 0
    prints the merged table, rank 1 prints nothing."""
    cfgfile = str(tmp_path / "rc.json")
    os.environ["DRAGNET_CONFIG"] = cfgfile
    from dragnet_amd import config as mod_config
    cfg = mod_config.DragnetConfig()
    cfg.datasource_add(mod_config.Datasource(
        name="shard_src", backend="sharded", path=fixture_tree))
    mod_config.save_config(cfg, cfgfile)

    # expected output from a plain single-process scan
    cfg2 = mod_config.DragnetConfig()
    cfg2.datasource_add(mod_config.Datasource(
        name="shard_src", backend="file", path=fixture_tree))
    mod_config.save_config(cfg2, str(tmp_path / "rc2.json"))
    import io

    from dragnet_amd import cli
    os.environ["DRAGNET_CONFIG"] = str(tmp_path / "rc2.json")
    os.environ["DRAGNET_ENGINE"] = "cpu"
    buf = io.StringIO()
    old = sys.stdout
    sys.stdout = buf
    try:
        assert cli.main(["scan", "-b", "req.method", "shard_src"]) == 0
    finally:
        sys.stdout = old
    expected = buf.getvalue()

    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_cli,
                         args=(r, 2, 29533, cfgfile, out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = dict()
    for _ in range(2):
        rank, code, out = out_q.get(timeout=90)
        assert code == 0
        results[rank] = out
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert results[0] == expected
    assert results[1] == ""


def _worker_build(rank, world, port, fixture_tree, idx_root, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DRAGNET_ENGINE"] = "cpu"
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))

    import dragnet_amd.distributed as D
    D._PARTITION_ROWS = 2  # force the hash-partition all_to_all path

    from dragnet_amd.config import Datasource
    from dragnet_amd.datasource.sharded import ShardedDatasource
    ds = Datasource(name="t", backend="sharded", path=fixture_tree,
                    index_path=idx_root, time_field="time",
                    time_format="%Y/%m-%d")
    sd = ShardedDatasource(ds)
    metrics = [{"name": "requests", "filter": None,
                "breakdowns": [
                    {"name": "req.method", "field": "req.method"},
                    {"name": "res.statusCode",
                     "field": "res.statusCode"}]}]
    written = sd.build(metrics, interval="day")
    out_q.put((rank, written))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_sharded_build_partitioned(fixture_tree, tmp_path):
    """Distributed build: the dense merge gives every rank the full
    table (via the forced hash-partition all_to_all path) and the
    interval buckets are written round-robin across ranks; the built
    tree must answer queries identically to a raw scan."""
    idx_root = str(tmp_path / "idx")
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    world = 2
    procs = [ctx.Process(target=_worker_build,
                         args=(r, world, 29537, fixture_tree,
                               idx_root, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = dict(out_q.get(timeout=120) for _ in range(world))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # rank 0 reports the full tree; rank 1 reports None (nonroot)
    assert results[1] is None
    full = results[0]
    assert full and all(os.path.exists(f) for f in full)

    # distributed build == single-process build (query both trees;
    # a raw scan is NOT the oracle here: the build's __dn_ts stage
    # drops the fixture's bad-date/missing-time records, by design)
    from dragnet_amd.config import Datasource
    from dragnet_amd.datasource.file import FileDatasource
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load
    metrics = [{"name": "requests", "filter": None,
                "breakdowns": [
                    {"name": "req.method", "field": "req.method"},
                    {"name": "res.statusCode",
                     "field": "res.statusCode"}]}]
    local_idx = str(tmp_path / "idx_local")
    ds1 = Datasource(name="t", backend="file", path=fixture_tree,
                     index_path=local_idx, time_field="time",
                     time_format="%Y/%m-%d")
    fd1 = FileDatasource(ds1, engine=CpuEngine())
    local_written = fd1.build(metrics, interval="day")
    assert [os.path.basename(f) for f in local_written] == \
        [os.path.basename(f) for f in full]

    q = query_load(breakdown_specs="req.method,res.statusCode")
    ds2 = Datasource(name="t", backend="file", path=fixture_tree,
                     index_path=idx_root, time_field="time",
                     time_format="%Y/%m-%d")
    fd2 = FileDatasource(ds2, engine=CpuEngine())
    via_dist = fd2.query(q, interval="day").aggregators[0].points()
    via_local = fd1.query(q, interval="day").aggregators[0].points()
    assert via_dist == via_local
    assert via_dist  # non-empty


@pytest.mark.timeout(180)
def test_launch_job_builder(fixture_tree, tmp_path):
    """dn-launch (the Manta job-builder analog): --dry-run prints the
    job definition; --gpus 2 actually runs a 2-rank distributed scan
    through torch.distributed.run whose rank-0 output equals a
    single-process scan."""
    import json as _json
    import subprocess
    import sys as _sys

    r = subprocess.run(
        [_sys.executable, "-m", "dragnet_amd.launch", "--gpus", "2",
         "--dry-run", "scan", "-b", "req.method", "src"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    job = _json.loads(r.stdout)
    assert job["nprocPerNode"] == 2
    assert job["phases"][0]["exec"].endswith(
        "-m dragnet_amd.cli scan -b req.method src")

    # real 2-rank run (gloo on CPU), output == single-process scan
    cfgfile = str(tmp_path / "rc.json")
    env = dict(os.environ, DRAGNET_CONFIG=cfgfile,
               DRAGNET_ENGINE="cpu")
    from dragnet_amd import config as mod_config
    cfg = mod_config.DragnetConfig()
    cfg.datasource_add(mod_config.Datasource(
        name="src", backend="sharded", path=fixture_tree))
    mod_config.save_config(cfg, cfgfile)
    r2 = subprocess.run(
        [_sys.executable, "-m", "dragnet_amd.launch", "--gpus", "2",
         "scan", "-b", "req.method", "src"],
        capture_output=True, text=True, env=env, timeout=150)
    assert r2.returncode == 0, r2.stderr[-2000:]

    cfg2 = mod_config.DragnetConfig()
    cfg2.datasource_add(mod_config.Datasource(
        name="src", backend="file", path=fixture_tree))
    mod_config.save_config(cfg2, str(tmp_path / "rc2.json"))
    env1 = dict(env, DRAGNET_CONFIG=str(tmp_path / "rc2.json"))
    r1_ = subprocess.run(
        [_sys.executable, "-m", "dragnet_amd.cli",
         "scan", "-b", "req.method", "src"],
        capture_output=True, text=True, env=env1)
    assert r1_.returncode == 0, r1_.stderr
    # gloo prints connection banners on stdout; drop them
    dist_out = "".join(
        ln for ln in r2.stdout.splitlines(keepends=True)
        if not ln.startswith("[Gloo]"))
    assert dist_out == r1_.stdout


def _worker_w4(rank, world, port, files, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["DRAGNET_ENGINE"] = "cpu"
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from dragnet_amd.distributed import (init_process_group,
                                         merge_tables_tensor)
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load
    init_process_group(backend="gloo")
    # rank 3 gets NO files (empty-shard edge); others round-robin 3-way
    mine = [] if rank == 3 else \
        [f for i, f in enumerate(files) if i % 3 == rank]
    q1 = query_load(breakdown_specs="operation,res.statusCode")
    q2 = query_load(filter={"eq": ["req.method", "NOSUCH"]},
                    breakdown_specs="host")  # matches nothing anywhere
    res = CpuEngine().scan(mine, [q1, q2])
    m1 = merge_tables_tensor(res.aggregators[0], q1)
    m2 = merge_tables_tensor(res.aggregators[1], q2)
    out_q.put((rank, m1.points(), m2.points()))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_world4_empty_shard_and_empty_tables(fixture_tree):
    """Dense merge at world 4 with one EMPTY rank shard and a second
    query whose filter matches nothing on any rank (all-empty
    all_gather): every rank must converge on the single-process
    answer."""
    files = []
    for root, _dirs, names in os.walk(fixture_tree):
        for n in sorted(names):
            files.append(os.path.join(root, n))
    files.sort()
    from dragnet_amd.engine.cpu import CpuEngine
    from dragnet_amd.query import query_load
    q1 = query_load(breakdown_specs="operation,res.statusCode")
    single = CpuEngine().scan(files, [q1])
    expected = single.aggregators[0].points()

    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    world = 4
    procs = [ctx.Process(target=_worker_w4,
                         args=(r, world, 29539, files, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [out_q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    for _rank, p1, p2 in results:
        assert p1 == expected
        assert p2 == []  # empty table merges to empty everywhere


def _worker_floatw(rank, world, port, out_q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from dragnet_amd.distributed import (init_process_group,
                                         merge_tables_tensor)
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    init_process_group(backend="gloo")
    q = query_load(breakdown_specs="k,lat[aggr=quantize]")
    agg = Aggregator(q)
    # float weights (json-skinner reduce shape), negative ordinals,
    # a shared key and rank-unique keys
    agg.table[("shared", 3)] = 0.5 + rank        # 0.5 / 1.5
    agg.table[("r%d" % rank, -2)] = 1.25
    agg.table[("int", 1)] = 2
    merged = merge_tables_tensor(agg, q)
    out_q.put((rank, dict(merged.table)))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_merge_float_weights_and_negative_ordinals():
    """Dense merge wire format: non-integer f64 values survive
    exactly, negative ordinals round-trip, int-typed totals stay
    ints."""
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker_floatw, args=(r, 2, 29541,
                                                      out_q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [out_q.get(timeout=90) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    expected = {("shared", 3): 2.0, ("r0", -2): 1.25,
                ("r1", -2): 1.25, ("int", 1): 4}
    for _rank, table in results:
        assert table == expected
        assert isinstance(table[("int", 1)], int)
        assert isinstance(table[("r0", -2)], float)


def test_wire_encode_huge_lquantize_ordinal():
    """lquantize ordinals beyond int64 (floor(1e300/step) is a
    ~1000-bit Python int) must survive the dense wire encode as f64
    keys instead of raising OverflowError (single-process slice of
    the C1 format)."""
    from dragnet_amd.distributed import _encode_table, _rebuild_table
    from dragnet_amd.points import Aggregator
    from dragnet_amd.query import query_load
    q = query_load(breakdown_specs="x[aggr=lquantize,step=7]")
    a = Aggregator(q)
    assert a.write({"fields": {"x": 1e300}, "value": 3})
    assert a.write({"fields": {"x": -1e300}, "value": 4})
    assert a.write({"fields": {"x": 2}, "value": 5})
    codes, tags, vals, strings = _encode_table(a, 1)
    out = _rebuild_table(q, codes, tags, vals, strings)
    assert sum(out.table.values()) == 12
    assert (0,) in out.table and out.table[(0,)] == 5
    huge = [k for k, in out.table if isinstance(k, float)]
    assert len(huge) == 2 and huge[0] != huge[1]

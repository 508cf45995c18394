"""Multi-process data-parallel tests on CPU (gloo, world_size=2): the same
distributed path the driver runs on 8 GPUs over RCCL, exercised here with
the gloo backend (SURVEY.md §4 flags the reference's lack of a loopback
multi-rank harness as a gap to fill)."""

import os

import torch
import torch.multiprocessing as mp


def _worker(rank, ws, store_path, q):
    import torch.distributed as dist

    from caffeonspark_amd.core import solver_from_prototxt
    from caffeonspark_amd.parallel import DistributedSync

    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        s = solver_from_prototxt(
            os.path.join(os.path.dirname(__file__), "..",
                         "caffeonspark_amd", "models",
                         "lenet_memory_solver.prototxt"))
        sync = DistributedSync(s, bucket_mb=0.5)
        sync.broadcast_params()
        g = torch.Generator().manual_seed(100 + rank)
        for it in range(5):
            x = torch.randn(64, 1, 28, 28, generator=g)
            y = torch.randint(0, 10, (64,), generator=g).float()
            s.net.data_layers()[0].reset(x, y)
            s._step_one()
        # plain numpy: tensor fd-sharing dies with the child process
        q.put((rank, s.flat_w.numpy().copy(), s.flat_g.numpy().copy()))
    finally:
        dist.destroy_process_group()


def _retry(fn, attempts=3):
    """gloo rendezvous is occasionally flaky under load; retry."""
    last = None
    for _ in range(attempts):
        try:
            return fn()
        except Exception as e:  # pragma: no cover - flake path
            last = e
    raise last


def test_ddp_two_ranks_identical_params():
    """After synchronized steps on different data, all ranks must hold
    bit-identical parameters."""
    _retry(_run_two_ranks_identical)


def _run_two_ranks_identical():
    ws = 2
    import tempfile, uuid
    store = tempfile.mktemp(prefix=f"cosamd_ddp_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, ws, store, q))
             for r in range(ws)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(ws):
        rank, w, gr = q.get(timeout=300)
        results[rank] = (w, gr)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    w0, g0 = results[0]
    w1, g1 = results[1]
    torch.testing.assert_close(torch.from_numpy(w0), torch.from_numpy(w1),
                               rtol=0, atol=0)
    torch.testing.assert_close(torch.from_numpy(g0), torch.from_numpy(g1),
                               rtol=1e-6, atol=1e-7)


def _seeded_solver():
    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    root = os.path.join(os.path.dirname(__file__), "..")
    sp = text_format.parse_file(
        os.path.join(root, "caffeonspark_amd", "models",
                     "lenet_memory_solver.prototxt"),
        caffe_pb.SolverParameter)
    sp.random_seed = 7
    sp.display = 0
    return Solver(sp, proto_dir=os.path.join(root, "caffeonspark_amd",
                                             "models"))


def _worker_equiv(rank, ws, store_path, q, data):
    import torch.distributed as dist

    from caffeonspark_amd.parallel import DistributedSync

    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        s = _seeded_solver()
        sync = DistributedSync(s)
        sync.broadcast_params()
        x, y = data
        half = x.shape[0] // ws
        s.net.data_layers()[0].reset(x[rank * half:(rank + 1) * half],
                                     y[rank * half:(rank + 1) * half])
        s._step_one()
        q.put((rank, s.flat_w.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_ddp_matches_single_process():
    """2-rank DP on split batch == single process on the full batch
    (gradient averaging equivalence, equal per-rank batch sizes)."""
    _retry(_run_matches_single)


def _run_matches_single():
    torch.manual_seed(5)
    x = torch.randn(128, 1, 28, 28)
    y = torch.randint(0, 10, (128,)).float()

    ws = 2
    import tempfile, uuid
    store = tempfile.mktemp(prefix=f"cosamd_ddp_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_equiv,
                         args=(r, ws, store, q, (x, y))) for r in range(ws)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(ws):
        rank, w = q.get(timeout=300)
        results[rank] = w
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    torch.testing.assert_close(torch.from_numpy(results[0]),
                               torch.from_numpy(results[1]),
                               rtol=0, atol=0)

    # single-process run over the whole batch from the same seeded init:
    # DP-averaged update must match the full-batch update (both losses are
    # per-sample means over equal halves)
    single = _seeded_solver()
    single.net.data_layers()[0].reset(x, y)
    single._step_one()
    torch.testing.assert_close(torch.from_numpy(results[0]), single.flat_w,
                               rtol=1e-4, atol=1e-6)


def test_bucket_assembly_covers_all_params():
    from caffeonspark_amd.core import solver_from_prototxt
    from caffeonspark_amd.parallel.ddp import DistributedSync

    s = solver_from_prototxt(
        os.path.join(os.path.dirname(__file__), "..",
                     "caffeonspark_amd", "models",
                     "lenet_memory_solver.prototxt"))
    sync = DistributedSync(s, bucket_mb=0.25)
    covered = set()
    for lo, hi in sync.buckets:
        covered.update(range(lo, hi, 1))
    total = int(s.flat_g.numel())
    # every param element inside some bucket
    for b, off in zip(s.params, s.param_offsets):
        assert off in covered and (off + b.count - 1) in covered
    assert len(sync.layer_bucket) == len(s.layer_slices)


def _worker_frozen(rank, ws, store_path, q):
    import torch.distributed as dist

    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.parallel import DistributedSync
    from caffeonspark_amd.proto import caffe_pb, text_format

    net_text = """
      layer { name: "d" type: "MemoryData" top: "x" top: "t"
              memory_data_param { batch_size: 32 channels: 3 height: 8
                                  width: 8 } }
      layer { name: "c" type: "Convolution" bottom: "x" top: "y"
              param { lr_mult: 0.0 } param { lr_mult: 0.0 }
              convolution_param { num_output: 8 kernel_size: 3 pad: 1
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "ip" type: "InnerProduct" bottom: "y" top: "z"
              inner_product_param { num_output: 4
                weight_filler { type: "gaussian" std: 0.1 } } }
      layer { name: "l" type: "SoftmaxWithLoss" bottom: "z" bottom: "t"
              top: "loss" }
    """
    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        sp = caffe_pb.SolverParameter(
            net_param=text_format.parse(net_text, caffe_pb.NetParameter),
            base_lr=0.1, momentum=0.9, lr_policy="fixed", max_iter=10,
            random_seed=5 + rank)   # deliberately different init per rank
        s = Solver(sp)
        sync = DistributedSync(s, bucket_mb=0.001)  # many small buckets
        sync.broadcast_params()
        frozen0 = s.params[0].data.clone()
        g = torch.Generator().manual_seed(200 + rank)
        for _ in range(4):
            x = torch.randn(32, 3, 8, 8, generator=g)
            y = torch.randint(0, 4, (32,), generator=g).float()
            s.net.data_layers()[0].reset(x, y)
            s._step_one()
        q.put((rank, s.flat_w.numpy().copy(),
               frozen0.numpy().copy(),
               s.params[0].data.detach().numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_ddp_frozen_params_stay_synchronized():
    """lr_mult=0 params inside all-reduce buckets: ranks stay bit-identical
    and the frozen blob never moves from its broadcast value."""
    _retry(_run_frozen)


def _run_frozen():
    import tempfile
    import uuid
    ws = 2
    store = tempfile.mktemp(prefix=f"cosamd_ddpf_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_frozen, args=(r, ws, store, q))
             for r in range(ws)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(ws):
        rank, flat_w, frozen0, frozen1 = q.get()
        out[rank] = (flat_w, frozen0, frozen1)
    for p in procs:
        p.join(timeout=60)
    torch.testing.assert_close(torch.from_numpy(out[0][0]),
                               torch.from_numpy(out[1][0]),
                               rtol=0, atol=0)
    for r in range(ws):
        torch.testing.assert_close(torch.from_numpy(out[r][1]),
                                   torch.from_numpy(out[r][2]),
                                   rtol=0, atol=0)


SHARED_NET = """
  layer { name: "d" type: "MemoryData" top: "x" top: "t"
          memory_data_param { batch_size: 16 channels: 1 height: 8
                              width: 8 } }
  layer { name: "ip1" type: "InnerProduct" bottom: "x" top: "h1"
          param { name: "shared_w" } param { name: "shared_b" }
          inner_product_param { num_output: 64
            weight_filler { type: "gaussian" std: 0.1 } } }
  layer { name: "r1" type: "ReLU" bottom: "h1" top: "h1" }
  layer { name: "ip2" type: "InnerProduct" bottom: "h1" top: "h2"
          inner_product_param { num_output: 64
            weight_filler { type: "gaussian" std: 0.1 } } }
  layer { name: "r2" type: "ReLU" bottom: "h2" top: "h2" }
  layer { name: "ip3" type: "InnerProduct" bottom: "h2" top: "h3"
          param { name: "shared_w" } param { name: "shared_b" }
          inner_product_param { num_output: 64
            weight_filler { type: "gaussian" std: 0.1 } } }
  layer { name: "ip4" type: "InnerProduct" bottom: "h3" top: "z"
          inner_product_param { num_output: 4
            weight_filler { type: "gaussian" std: 0.1 } } }
  layer { name: "l" type: "SoftmaxWithLoss" bottom: "z" bottom: "t"
          top: "loss" }
"""


def _shared_solver(seed=7):
    from caffeonspark_amd.core.solver import Solver
    from caffeonspark_amd.proto import caffe_pb, text_format

    sp = caffe_pb.SolverParameter(
        net_param=text_format.parse(SHARED_NET, caffe_pb.NetParameter),
        base_lr=0.05, momentum=0.9, lr_policy="fixed", max_iter=10,
        random_seed=seed, display=0)
    return Solver(sp)


def test_shared_param_buckets_exactly_once():
    """caffe named-param sharing (param { name }): two layers alias one
    arena range. Buckets must partition the arena — the old per-layer
    min/max span assembly emitted overlapping buckets that all-reduced
    the shared range twice (x world_size gradient inflation) and raced
    two async all_reduce calls on the same buffer."""
    from caffeonspark_amd.parallel.ddp import DistributedSync

    s = _shared_solver()
    sync = DistributedSync(s, bucket_mb=0.01)  # force several buckets
    assert len(sync.buckets) >= 2
    hits = torch.zeros(int(s.flat_g.numel()), dtype=torch.int32)
    for lo, hi in sync.buckets:
        hits[lo:hi] += 1
    assert int(hits.min()) == 1 and int(hits.max()) == 1

    # the shared blob's bucket fires only at its EARLIEST layer's backward
    # (the last of its consumers to run in reverse order)
    shared_off = s.param_offsets[0]  # ip1_w is first param in the arena
    shared_bucket = next(i for i, (lo, hi) in enumerate(sync.buckets)
                         if lo <= shared_off < hi)
    assert sync.layer_bucket.get("ip1") == shared_bucket
    assert "ip3" not in sync.layer_bucket or \
        sync.layer_bucket["ip3"] != shared_bucket or \
        sync.layer_bucket.get("ip1") == sync.layer_bucket.get("ip3")


def _worker_shared(rank, ws, store_path, q, data):
    import torch.distributed as dist

    from caffeonspark_amd.parallel import DistributedSync

    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        s = _shared_solver()
        sync = DistributedSync(s, bucket_mb=0.02)
        sync.broadcast_params()
        x, y = data
        half = x.shape[1] // ws
        for it in range(3):
            s.net.data_layers()[0].reset(
                x[it][rank * half:(rank + 1) * half],
                y[it][rank * half:(rank + 1) * half])
            s._step_one()
        q.put((rank, s.flat_w.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_ddp_shared_params_match_single():
    """2-rank DP with param{name} weight sharing == single process on the
    full batch. The old overlapping-bucket assembly double-reduced the
    shared range and diverged."""
    _retry(_run_shared_match)


def _run_shared_match():
    import tempfile, uuid
    torch.manual_seed(11)
    x = torch.stack([torch.randn(32, 1, 8, 8) for _ in range(3)])
    y = torch.stack([torch.randint(0, 4, (32,)).float() for _ in range(3)])
    ws = 2
    store = tempfile.mktemp(prefix=f"cosamd_ddps_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_shared,
                         args=(r, ws, store, q, (x, y))) for r in range(ws)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(ws):
        rank, w = q.get(timeout=300)
        results[rank] = w
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    torch.testing.assert_close(torch.from_numpy(results[0]),
                               torch.from_numpy(results[1]),
                               rtol=0, atol=0)
    single = _shared_solver()
    for it in range(3):
        single.net.data_layers()[0].reset(x[it], y[it])
        single._step_one()
    torch.testing.assert_close(torch.from_numpy(results[0]), single.flat_w,
                               rtol=1e-4, atol=1e-6)


def _worker_lrcn(rank, ws, store_path, q):
    import torch.distributed as dist

    from caffeonspark_amd.core import solver_from_prototxt
    from caffeonspark_amd.core.layers.data import CoSDataLayer
    from caffeonspark_amd.parallel import DistributedSync

    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        s = solver_from_prototxt(
            os.path.join(os.path.dirname(__file__), "..",
                         "caffeonspark_amd", "models",
                         "lrcn_solver.prototxt"))
        s.param.display = 0
        sync = DistributedSync(s, bucket_mb=4.0)
        sync.broadcast_params()
        dl = s.net.data_layers()[0]
        assert isinstance(dl, CoSDataLayer)
        T = int(dl.tops_cfg[2].channels)
        V, n = 8801, 2
        g = torch.Generator().manual_seed(50 + rank)
        dl.batch_size = n
        for _ in range(2):
            x = torch.randn(n, 3, 227, 227, generator=g)
            cont = torch.ones(T, n)
            cont[0] = 0
            inp = torch.randint(0, V, (T, n), generator=g).float()
            tgt = torch.randint(0, V, (T, n), generator=g).float()
            dl.reset([x, torch.zeros(n, 1), cont, inp, tgt])
            s._step_one()
        q.put((rank, s.flat_w.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_ddp_lrcn_two_ranks_identical():
    """The actual LRCN net (CNN+LSTM, the BASELINE 4-GPU config) under
    2-rank gloo DP: ranks stay bit-identical through synchronized steps
    on different data."""
    _retry(_run_lrcn)


def _run_lrcn():
    import tempfile, uuid
    ws = 2
    store = tempfile.mktemp(prefix=f"cosamd_lrcn_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_lrcn, args=(r, ws, store, q))
             for r in range(ws)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(ws):
        rank, w = q.get(timeout=600)
        results[rank] = w
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    torch.testing.assert_close(torch.from_numpy(results[0]),
                               torch.from_numpy(results[1]),
                               rtol=0, atol=0)


def _worker_bf16wire(rank, ws, store_path, q, data):
    import os

    import torch.distributed as dist

    from caffeonspark_amd.parallel import DistributedSync

    os.environ["COS_DDP_BF16"] = "1"     # force the bf16 wire on gloo
    dist.init_process_group("gloo", rank=rank, world_size=ws,
                            init_method=f"file://{store_path}")
    try:
        s = _seeded_solver()
        sync = DistributedSync(s)
        sync.broadcast_params()
        x, y = data
        half = x.shape[0] // ws
        s.net.data_layers()[0].reset(x[rank * half:(rank + 1) * half],
                                     y[rank * half:(rank + 1) * half])
        s._step_one()
        q.put((rank, s.flat_w.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_ddp_bf16_gradient_wire():
    """bf16 gradient all-reduce (COS_DDP_BF16=1): ranks stay identical
    and land within bf16 tolerance of the fp32-wire full-batch result."""
    _retry(_run_bf16_wire)


def _run_bf16_wire():
    import tempfile
    import uuid
    torch.manual_seed(5)
    x = torch.randn(128, 1, 28, 28)
    y = torch.randint(0, 10, (128,)).float()
    ws = 2
    store = tempfile.mktemp(prefix=f"cosamd_bf16_{uuid.uuid4().hex}_")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_bf16wire,
                         args=(r, ws, store, q, (x, y))) for r in range(ws)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(ws):
        rank, w = q.get(timeout=300)
        results[rank] = w
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    torch.testing.assert_close(torch.from_numpy(results[0]),
                               torch.from_numpy(results[1]),
                               rtol=0, atol=0)
    single = _seeded_solver()
    single.net.data_layers()[0].reset(x, y)
    single._step_one()
    # bf16 wire: ~3 decimal digits on the gradient; one SGD step keeps
    # weights within bf16-grad * lr of the fp32 path
    torch.testing.assert_close(torch.from_numpy(results[0]), single.flat_w,
                               rtol=5e-3, atol=5e-4)

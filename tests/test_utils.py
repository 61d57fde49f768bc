"""Tests for checkpoint / metrics / data / profiler utilities."""

import torch

from geomx_amd import Config
from geomx_amd.kvstore.optimizer import OptimizerSpec
from geomx_amd.models import geo_cnn
from geomx_amd.parallel import GeoTrainer
from geomx_amd.topology import init_topology
from geomx_amd.utils import checkpoint as ckpt
from geomx_amd.utils.data import (ClassSplitSampler, SplitSampler,
                                  SyntheticImageDataset, worker_loader)
from geomx_amd.utils.metrics import Accuracy, Measure


def test_save_load_parameters(tmp_path):
    m1 = torch.nn.Linear(4, 3)
    f = str(tmp_path / "m.params")
    ckpt.save_parameters(m1, f)
    m2 = torch.nn.Linear(4, 3)
    ckpt.load_parameters(m2, f)
    for a, b in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(a, b)


def test_checkpoint_roundtrip_with_trainer(tmp_path):
    torch.manual_seed(0)
    cfg = Config.from_env(device="cpu")
    topo = init_topology(1, None, "gloo", "cpu")
    model = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 2))
    tr = GeoTrainer(model, cfg, topo, OptimizerSpec("adam", lr=0.01))
    x = torch.randn(4, 8)
    for _ in range(2):
        loss = model(x).square().mean()
        tr.zero_grad()
        loss.backward()
        tr.step()
    files = ckpt.save_checkpoint(model, tr, str(tmp_path / "ck"), 1)
    w_before = [p.detach().clone() for p in model.parameters()]

    # new model+trainer, resumed
    torch.manual_seed(123)
    model2 = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 2))
    tr2 = GeoTrainer(model2, cfg, topo, OptimizerSpec("adam", lr=0.01))
    ckpt.load_checkpoint(model2, tr2, str(tmp_path / "ck"), 1)
    for a, b in zip(w_before, model2.parameters()):
        assert torch.allclose(a, b, atol=1e-7)
    # optimizer state came back
    assert tr2.server_opt.step_count == tr.server_opt.step_count
    # training continues identically
    for t in (tr, tr2):
        pass
    loss1 = model(x).square().mean()
    tr.zero_grad(); loss1.backward(); tr.step()
    loss2 = model2(x).square().mean()
    tr2.zero_grad(); loss2.backward(); tr2.step()
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.allclose(a, b, atol=1e-6)


def test_split_sampler_partitions():
    s0 = SplitSampler(100, 4, 0, shuffle=False)
    s3 = SplitSampler(100, 4, 3, shuffle=False)
    assert list(s0) == list(range(25))
    assert list(s3) == list(range(75, 100))
    all_idx = []
    for i in range(4):
        all_idx += list(SplitSampler(100, 4, i, shuffle=False))
    assert sorted(all_idx) == list(range(100))


def test_class_split_sampler_disjoint():
    ds = SyntheticImageDataset(200, shape=(1, 8, 8), num_classes=10)
    seen = [set(ds.y[list(ClassSplitSampler(ds.y, 2, i))].tolist())
            for i in range(2)]
    assert seen[0].isdisjoint(seen[1])
    assert seen[0] | seen[1] == set(range(10))


def test_worker_loader_batches():
    ds = SyntheticImageDataset(128, shape=(3, 8, 8))
    dl = worker_loader(ds, 16, 2, 0)
    xb, yb = next(iter(dl))
    assert xb.shape == (16, 3, 8, 8)


def test_measure_and_accuracy(tmp_path):
    m = Measure(sync_cuda=False)
    m.start("fwd")
    m.stop("fwd")
    row = m.next_iteration(iter=1)
    assert "fwd" in row and row["fwd"] >= 0
    f = str(tmp_path / "meas.jsonl")
    m.dump(f)
    assert open(f).read().strip()

    acc = Accuracy()
    acc.update(torch.tensor([1, 2]), torch.tensor([[0, 1.0, 0], [0, 0, 1.0]]))
    assert acc.get() == 1.0


def test_example_convergence_cpu():
    """Convergence smoke (the reference's de-facto system test is
    per-iteration accuracy climbing, cnn.py:129-131): single-process
    vanilla kvstore training on learnable synthetic data."""
    from geomx_amd.kvstore import create
    from geomx_amd.utils.metrics import eval_acc

    cfg = Config.from_env(device="cpu")
    kv = create("dist_sync", cfg=cfg)
    kv.set_optimizer(OptimizerSpec("adam", lr=0.003))
    torch.manual_seed(0)
    net = geo_cnn(in_channels=3, image_size=16)
    ds = SyntheticImageDataset(512, shape=(3, 16, 16), seed=1)
    dl = torch.utils.data.DataLoader(ds, batch_size=64, shuffle=True)
    test = torch.utils.data.DataLoader(
        SyntheticImageDataset(128, shape=(3, 16, 16), seed=1), batch_size=64)
    params = [p for p in net.parameters()]
    for i, p in enumerate(params):
        kv.init(i, p.data)
    acc0 = eval_acc(net, test, "cpu")
    for epoch in range(3):
        for x, y in dl:
            loss = torch.nn.functional.cross_entropy(net(x), y)
            net.zero_grad()
            loss.backward()
            for i, p in enumerate(params):
                kv.push(i, p.grad / x.shape[0])
                kv.pull(i, p.data)
    acc1 = eval_acc(net, test, "cpu")
    assert acc1 > max(0.3, acc0 + 0.15), (acc0, acc1)


def test_heartbeat_monitor_single():
    import time

    from geomx_amd.utils.health import HeartbeatMonitor

    class FakeStore:
        def __init__(self):
            self.d = {}

        def set(self, k, v):
            self.d[k] = v

        def get(self, k):
            return self.d[k]

    st = FakeStore()
    hb = HeartbeatMonitor(interval_s=0.05, store=st, rank=0, world_size=2)
    hb.start()
    time.sleep(0.2)
    # rank 0 alive, rank 1 never beat
    assert hb.dead_nodes(timeout_s=10.0) == [1]
    assert hb.get_num_dead_node(10.0) == 1
    hb.stop()


def test_profiler_wrapper(tmp_path):
    from geomx_amd.utils.profiler import Profiler
    p = Profiler(filename=str(tmp_path / "trace.json"))
    p.set_state("run")
    x = torch.randn(64, 64)
    (x @ x).sum()
    p.set_state("stop")
    out = p.dump(rank=3)
    assert out and out.endswith("rank3_trace.json")
    import os
    assert os.path.exists(out)


def test_server_profiler_command_single():
    from geomx_amd import Config
    from geomx_amd.kvstore import create
    from geomx_amd.utils import profiler as prof

    kv = create("dist_sync", cfg=Config.from_env(device="cpu"))
    prof.send_server_profiler_command(kv, prof.ServerProfilerCommand.STATE,
                                      "run")
    prof.send_server_profiler_command(kv, prof.ServerProfilerCommand.PAUSE)
    out = prof.send_server_profiler_command(
        kv, prof.ServerProfilerCommand.DUMP)
    prof._default.set_state("stop")


# ---------------------------------------------------------------------------
# recordio: the im2rec / RecordIO analog (tools/im2rec.py, dmlc RecordIO)
# ---------------------------------------------------------------------------

def test_recordio_roundtrip(tmp_path):
    from geomx_amd.utils.data import SyntheticImageDataset
    from geomx_amd.utils.recordio import RecordDataset, pack_dataset

    ds = SyntheticImageDataset(n=12, shape=(3, 8, 8), num_classes=4)
    path = str(tmp_path / "train.rec")
    n = pack_dataset(ds, path)
    assert n == 12

    rd = RecordDataset(path)
    assert len(rd) == 12
    for i in [0, 7, 11, 3]:  # random access order
        x, y = rd[i]
        xe, ye = ds[i]
        assert torch.equal(x, xe) and y == int(ye)
    assert torch.equal(rd.labels, ds.y)


def test_recordio_dtypes_and_scalar(tmp_path):
    from geomx_amd.utils.recordio import RecordDataset, RecordWriter

    path = str(tmp_path / "mixed.rec")
    vals = [torch.arange(6, dtype=torch.int32).reshape(2, 3),
            torch.randn(5).to(torch.bfloat16),
            torch.tensor(3.5),                       # 0-dim
            (torch.rand(4, 4) * 255).to(torch.uint8)]
    with RecordWriter(path) as w:
        for i, v in enumerate(vals):
            w.write(v, label=i * 10)
    rd = RecordDataset(path)
    for i, v in enumerate(vals):
        x, y = rd[i]
        assert x.dtype == v.dtype and x.shape == v.shape
        assert torch.equal(x, v)
        assert y == i * 10


def test_recordio_dataloader_workers(tmp_path):
    from geomx_amd.utils.data import SplitSampler, SyntheticImageDataset
    from geomx_amd.utils.recordio import RecordDataset, pack_dataset

    ds = SyntheticImageDataset(n=16, shape=(2, 4, 4), num_classes=2)
    path = str(tmp_path / "w.rec")
    pack_dataset(ds, path)
    rd = RecordDataset(path)
    sampler = SplitSampler(len(rd), num_parts=2, part_index=0, shuffle=False)
    dl = torch.utils.data.DataLoader(rd, batch_size=4, sampler=sampler,
                                     num_workers=2)
    seen = 0
    for xb, yb in dl:
        assert xb.shape == (4, 2, 4, 4)
        seen += xb.shape[0]
    assert seen == 8  # worker 0's half


def test_recordio_corruption_detected(tmp_path):
    import pytest
    from geomx_amd.utils.recordio import RecordDataset, RecordWriter

    path = str(tmp_path / "c.rec")
    with RecordWriter(path) as w:
        w.write(torch.ones(3), label=1)
    with open(path, "r+b") as f:
        f.seek(0)
        f.write(b"\x00\x00\x00\x00")  # clobber magic
    # both readers must refuse the corrupt frame (native raises
    # RuntimeError via pybind, the python path raises IOError)
    with pytest.raises((IOError, RuntimeError)):
        RecordDataset(path)[0]
    with pytest.raises(IOError):
        RecordDataset(path, native=False)[0]


def test_party_wan_gbps_config(monkeypatch):
    """Heterogeneous per-party WAN rates: env parsing, validation, and
    rate lookup."""
    import pytest
    from geomx_amd import Config

    monkeypatch.setenv("GEOMX_PARTY_WAN_GBPS", "1.0,0.25")
    cfg = Config.from_env(num_parties=2)
    assert cfg.party_wan_gbps == [1.0, 0.25]
    assert cfg.wan_rate_for(0) == 1.0 and cfg.wan_rate_for(1) == 0.25
    monkeypatch.delenv("GEOMX_PARTY_WAN_GBPS")

    cfg2 = Config.from_env(num_parties=2, wan_gbps=3.0)
    assert cfg2.wan_rate_for(0) == 3.0 == cfg2.wan_rate_for(1)

    with pytest.raises(ValueError):
        Config.from_env(num_parties=3,
                        party_wan_gbps=[1.0, 2.0]).validate()


def test_recordio_native_matches_python(tmp_path):
    """The C++ reader (_geoio) and the python mmap path return
    identical tensors/labels; read_batch assembles correctly."""
    import pytest
    from geomx_amd.utils.data import SyntheticImageDataset
    from geomx_amd.utils.recordio import (RecordDataset, _geoio,
                                          pack_dataset)
    if _geoio is None:
        pytest.skip("_geoio not built")

    ds = SyntheticImageDataset(n=10, shape=(3, 6, 6), num_classes=3)
    path = str(tmp_path / "n.rec")
    pack_dataset(ds, path)
    nat = RecordDataset(path, native=True)
    py = RecordDataset(path, native=False)
    assert len(nat) == len(py) == 10
    for i in range(10):
        xn, yn = nat[i]
        xp, yp = py[i]
        assert torch.equal(xn, xp) and yn == yp
    assert torch.equal(nat.labels, py.labels)

    idx = [3, 0, 7, 7, 1]
    bn, ln = nat.read_batch(idx, threads=3)
    bp, lp = py.read_batch(idx)
    assert torch.equal(bn, bp) and torch.equal(ln, lp)
    assert bn.shape == (5, 3, 6, 6)

    # mixed shapes in one batch must fail loudly, not corrupt
    from geomx_amd.utils.recordio import RecordWriter
    p2 = str(tmp_path / "mix.rec")
    with RecordWriter(p2) as w:
        w.write(torch.ones(4), 0)
        w.write(torch.ones(5), 1)
    with pytest.raises(RuntimeError):
        RecordDataset(p2, native=True).read_batch([0, 1])


def test_record_batch_loader(tmp_path):
    from geomx_amd.utils.data import SplitSampler, SyntheticImageDataset
    from geomx_amd.utils.recordio import (RecordBatchLoader, RecordDataset,
                                          pack_dataset)

    ds = SyntheticImageDataset(n=20, shape=(2, 4, 4), num_classes=3)
    path = str(tmp_path / "bl.rec")
    pack_dataset(ds, path)
    rd = RecordDataset(path)
    sampler = SplitSampler(len(rd), num_parts=2, part_index=1,
                           shuffle=False)
    dl = RecordBatchLoader(rd, batch_size=4, sampler=sampler)
    assert len(dl) == 3  # 10 records of worker 1 in batches of 4
    seen = 0
    for xb, yb in dl:
        for j in range(xb.shape[0]):
            i = 10 + seen + j  # worker 1's contiguous shard
            assert torch.equal(xb[j], ds.x[i])
            assert yb[j] == ds.y[i]
        seen += xb.shape[0]
    assert seen == 10

    # drop_last
    dl2 = RecordBatchLoader(rd, batch_size=3, drop_last=True)
    assert len(dl2) == 6
    assert sum(x.shape[0] for x, _ in dl2) == 18


def test_recordio_empty_file(tmp_path):
    from geomx_amd.utils.recordio import RecordDataset, RecordWriter

    path = str(tmp_path / "e.rec")
    with RecordWriter(path):
        pass
    for native in (None, False):
        rd = RecordDataset(path, native=native)
        assert len(rd) == 0

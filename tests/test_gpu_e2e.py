"""GPU end-to-end: flagship model steps through the trainer and kvstore
on one MI355X, exercising the native kernel path."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import geomx_amd.ops as ops
    assert ops.native_available(), ops.native_error()


def test_trainer_cnn_step():
    from geomx_amd import Config
    from geomx_amd.kvstore.optimizer import OptimizerSpec
    from geomx_amd.models import create_model
    from geomx_amd.parallel import GeoTrainer
    from geomx_amd.topology import init_topology

    dev = torch.device("cuda:0")
    cfg = Config.from_env()
    topo = init_topology(1, None, None, "cuda:0")
    model = create_model("geomx_cnn", image_size=64).to(dev)
    tr = GeoTrainer(model, cfg, topo, OptimizerSpec("adam", lr=1e-3))
    x = torch.randn(16, 3, 64, 64, device=dev)
    y = torch.randint(0, 10, (16,), device=dev)
    losses = []
    for _ in range(8):
        with torch.autocast("cuda", torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        tr.zero_grad()
        loss.backward()
        tr.step()
        losses.append(loss.item())
    # training on a fixed batch must reduce the loss
    assert losses[-1] < losses[0], losses


def test_kvstore_gpu_single():
    from geomx_amd import Config
    from geomx_amd.kvstore import create
    from geomx_amd.kvstore.optimizer import OptimizerSpec

    cfg = Config.from_env(device="cuda:0")
    kv = create("dist_sync", cfg=cfg)
    kv.set_optimizer(OptimizerSpec("sgd", lr=0.1))
    kv.init("w", torch.ones(1024, device="cuda:0"))
    kv.push("w", torch.full((1024,), 2.0, device="cuda:0"))
    out = torch.empty(1024, device="cuda:0")
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((1024,), 0.8, device="cuda:0"))


def test_kvstore_gpu_bsc_roundtrip():
    from geomx_amd import Config
    from geomx_amd.kvstore import create

    cfg = Config.from_env(device="cuda:0")
    kv = create("dist_sync", cfg=cfg)
    kv.set_gradient_compression({"type": "bsc", "threshold": 0.01})
    n = 1 << 20
    kv.init("w", torch.zeros(n, device="cuda:0"))
    g = torch.randn(n, device="cuda:0")
    kv.push("w", g)
    out = torch.empty(n, device="cuda:0")
    kv.pull("w", out)
    # single party: push is the identity (no WAN compression applied)
    assert torch.allclose(out, g)


def test_resnet50_smoke():
    from geomx_amd.models import create_model
    dev = torch.device("cuda:0")
    model = create_model("resnet50").to(dev)
    x = torch.randn(4, 3, 224, 224, device=dev)
    with torch.autocast("cuda", torch.bfloat16):
        y = model(x)
        loss = y.float().square().mean()
    loss.backward()
    assert torch.isfinite(loss).item()


def test_recordio_pinned_batch_to_gpu(tmp_path):
    """Native record reader feeding the GPU: threaded batch assembly
    into pinned host memory, then H2D — the intended data path on a
    training rank."""
    from geomx_amd.utils.data import SyntheticImageDataset
    from geomx_amd.utils.recordio import RecordDataset, _geoio, pack_dataset

    assert _geoio is not None, "_geoio extension not built"
    ds = SyntheticImageDataset(n=32, shape=(3, 16, 16), num_classes=4)
    path = str(tmp_path / "g.rec")
    pack_dataset(ds, path)
    rd = RecordDataset(path, native=True)
    xb, yb = rd.read_batch(range(16), threads=4, pin_memory=True)
    assert xb.is_pinned()
    dev = xb.to("cuda:0", non_blocking=True)
    torch.cuda.synchronize()
    assert torch.equal(dev.cpu(), ds.x[:16])
    assert torch.equal(yb, ds.y[:16])


def test_kvstore_gpu_row_sparse_and_dgt_payload():
    """r02 paths on GPU, single process: rows-only row_sparse_pull and
    the DGT mode-3 wire payload compress/decompress round-trip."""
    import geomx_amd
    from geomx_amd import Config
    from geomx_amd.kvstore import create
    from geomx_amd.kvstore.dgt import DGTState
    cfg = Config.from_env(device="cuda:0")
    kv = create("dist_sync", cfg=cfg)
    torch.manual_seed(2)
    w = torch.randn(64, 8, device=DEV)
    kv.init("emb", w)
    g = torch.randn(64, 8, device=DEV)
    kv.push("emb", g)
    dense = torch.empty(64, 8, device=DEV)
    kv.pull("emb", dense)
    ids = torch.tensor([3, 17, 3, 60], device=DEV)
    out = torch.empty(4, 8, device=DEV)
    kv.row_sparse_pull("emb", out, ids)
    assert torch.allclose(out, dense[ids.long()], atol=1e-6)

    st = DGTState(1 << 16, DEV, chunk_elems=256, k=0.25, mode=3)
    x = torch.randn(1 << 16, device=DEV)
    rec = st.decompress(*st.compress(x))
    st2 = DGTState(1 << 16, DEV, chunk_elems=256, k=0.25, mode=3)
    ref_out, _ = st2.transform(x.clone())
    assert torch.allclose(rec, ref_out, atol=1e-4)


def test_trainer_bsc_dgt_gpu_single():
    """bsc_dgt composed WAN tier runs on the GPU kernels (world 1:
    exercises the compress path through ops without a process group)."""
    from geomx_amd import ops
    n = 1 << 18
    g = torch.randn(n, device=DEV)
    u = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    vals, idx = ops.bsc_compress(g, u, v, 0.01)
    from geomx_amd.kvstore.dgt import DGTState
    dg = DGTState(vals.numel(), DEV, chunk_elems=256, k=0.5, mode=3)
    vals_z = vals.masked_fill(idx < 0, 0.0)
    rec = dg.decompress(*dg.compress(vals_z))
    dense = ops.bsc_decompress(rec.contiguous(), idx, n)
    assert torch.isfinite(dense).all()
    sel = int((idx >= 0).sum())
    assert dense.abs().sum() > 0 and sel > 0

"""Multi-process CPU (gloo) tests of the hierarchical kvstore.

These cover the distributed semantics the driver can check without a
GPU: FSA two-tier aggregation, update-on-server vs update-on-worker,
HFA, Bi-Sparse / FP16 / MPQ global-tier compression, MixedSync
sequential async updates, and the WAN token-bucket accounting.
"""

import pytest
import torch

from dist_helpers import run_dist

from geomx_amd import Config
from geomx_amd.kvstore import create
from geomx_amd.kvstore.optimizer import OptimizerSpec
from geomx_amd.ops import reference as ref
from geomx_amd.topology import init_topology


def _mk(mode="dist_sync", num_parties=1, **over):
    cfg = Config.from_env(num_parties=num_parties, backend="gloo",
                          device="cpu", **over)
    topo = init_topology(cfg.num_parties, cfg.party_sizes, "gloo", "cpu")
    return create(mode, cfg=cfg, topo=topo)


# ---------------------------------------------------------------------------
# FSA: flat sum semantics (1 party)
# ---------------------------------------------------------------------------

def _fsa_flat(rank, world):
    kv = _mk(num_parties=1)
    torch.manual_seed(0)  # same init on all ranks
    w = torch.randn(8)
    kv.init("w", w)
    g = torch.full((8,), float(rank + 1))
    kv.push("w", g)
    out = torch.empty(8)
    kv.pull("w", out)
    expected = sum(r + 1 for r in range(world))
    assert torch.allclose(out, torch.full((8,), float(expected))), out


def test_fsa_flat_sum_ws2():
    run_dist(2, _fsa_flat)


# ---------------------------------------------------------------------------
# FSA: hierarchical 2 parties x 2 workers
# ---------------------------------------------------------------------------

def _fsa_hier(rank, world, global_mode):
    kv = _mk(num_parties=2, **{})
    kv.global_mode = global_mode
    w = torch.zeros(6)
    kv.init("w", w)
    g = torch.full((6,), float(rank + 1))
    kv.push("w", g)
    out = torch.empty(6)
    kv.pull("w", out)
    expected = sum(r + 1 for r in range(world))  # sum over both tiers
    assert torch.allclose(out, torch.full((6,), float(expected))), \
        (rank, out, expected)


@pytest.mark.parametrize("global_mode", ["sharded", "replicated"])
def test_fsa_hierarchical_ws4(global_mode):
    run_dist(4, _fsa_hier, global_mode)


# ---------------------------------------------------------------------------
# update-on-server: global-server optimizer, workers pull params
# ---------------------------------------------------------------------------

def _server_sgd(rank, world):
    kv = _mk(num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    kv.init("w", torch.ones(5))
    kv.push("w", torch.full((5,), 1.0))   # sum over 4 workers = 4
    out = torch.empty(5)
    kv.pull("w", out)
    # w = 1 - 0.1*4 = 0.6 on every worker in every party
    assert torch.allclose(out, torch.full((5,), 0.6), atol=1e-6), (rank, out)
    # second step from the updated value
    kv.push("w", torch.full((5,), 0.5))   # sum = 2
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((5,), 0.4), atol=1e-6), (rank, out)


def test_update_on_server_sharded_ws4():
    run_dist(4, _server_sgd)


# ---------------------------------------------------------------------------
# Bi-Sparse over the WAN tier (update-on-worker)
# ---------------------------------------------------------------------------

def _bsc_push_pull(rank, world):
    kv = _mk(num_parties=2)
    # size_lower_bound below n: the reference gates SMALL tensors to a
    # plain push even on the bsc path (kvstore_dist_server.h:879)
    kv.set_gradient_compression({"type": "bsc", "threshold": 0.05,
                                 "size_lower_bound": 1000})
    n = 2000
    kv.init("w", torch.zeros(n))
    torch.manual_seed(7)  # identical grads on all ranks for predictability
    g = torch.randn(n)
    kv.push("w", g)
    out = torch.empty(n)
    kv.pull("w", out)
    # Aggregated gradient: each party's sum (2g) is bsc-compressed; with
    # identical inputs both parties send the same top-k values, so the
    # result is 2*sum_of_parties for sent coords = 4g at sent coords.
    sent = out != 0
    assert sent.sum() > 0
    assert torch.allclose(out[sent], 4 * g[sent], atol=1e-5)
    # sparsity: roughly capacity-bound (2 parties x k each, overlapping)
    k = ref.bsc_capacity(n, 0.05)
    assert sent.sum() <= 2 * k


def test_bsc_ws4():
    run_dist(4, _bsc_push_pull)


# ---------------------------------------------------------------------------
# FP16 global tier
# ---------------------------------------------------------------------------

def _fp16(rank, world):
    kv = _mk(num_parties=2)
    kv.set_gradient_compression({"type": "fp16"})
    kv.init("w", torch.zeros(64))
    g = torch.full((64,), 0.25)  # exactly representable in fp16
    kv.push("w", g)
    out = torch.empty(64)
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((64,), 0.25 * world), atol=1e-3)


def test_fp16_ws4():
    run_dist(4, _fp16)


# ---------------------------------------------------------------------------
# MPQ: size gate routes small->fp16, large->bsc
# ---------------------------------------------------------------------------

def _mpq(rank, world):
    kv = _mk(num_parties=2)
    kv.set_gradient_compression(
        {"type": "mpq", "threshold": 0.05, "size_lower_bound": 100})
    kv.init("small", torch.zeros(10))
    kv.init("large", torch.zeros(4000))
    kv.push("small", torch.full((10,), 0.5))
    torch.manual_seed(3)
    g = torch.randn(4000)
    kv.push("large", g)
    small = torch.empty(10)
    large = torch.empty(4000)
    kv.pull("small", small)
    kv.pull("large", large)
    # small went fp16 dense: exact sum
    assert torch.allclose(small, torch.full((10,), 0.5 * world), atol=1e-3)
    # large went bsc: sparse
    assert (large == 0).sum() > 2000


def test_mpq_ws4():
    run_dist(4, _mpq)


# ---------------------------------------------------------------------------
# HFA: K2 gating + milestone delta rebase (model averaging)
# ---------------------------------------------------------------------------

def _hfa(rank, world):
    kv = _mk(num_parties=2, use_hfa=True, hfa_k2=2)
    kv.init("w", torch.zeros(4))
    # HFA pushes param/num_local_workers (model averaging). Party 0
    # workers hold 2.0, party 1 workers hold 6.0.
    party = 0 if rank < 2 else 1
    local_param = torch.full((4,), 2.0 if party == 0 else 6.0)
    nloc = 2
    out = torch.empty(4)

    # push 1: no global sync (1 % K2 != 0) -> pull returns party average
    kv.push("w", local_param / nloc)
    kv.pull("w", out)
    assert torch.allclose(out, local_param), (rank, out)

    # push 2: global sync -> average across parties = 4.0
    kv.push("w", local_param / nloc)
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((4,), 4.0)), (rank, out)


def test_hfa_ws4():
    run_dist(4, _hfa)


# ---------------------------------------------------------------------------
# MixedSync (dist_async): sequential per-party optimizer updates
# ---------------------------------------------------------------------------

def _mixed_async(rank, world):
    kv = _mk("dist_async", num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    kv.init("w", torch.ones(3))
    kv.push("w", torch.full((3,), 1.0))
    out = torch.empty(3)
    kv.pull("w", out)
    # two sequential updates of party sums (2.0 each): w = 1 - .1*2 - .1*2
    assert torch.allclose(out, torch.full((3,), 0.6), atol=1e-6), (rank, out)


def test_mixed_async_ws4():
    run_dist(4, _mixed_async)


# ---------------------------------------------------------------------------
# DCASGD delay compensation runs under dist_async
# ---------------------------------------------------------------------------

def _dcasgd(rank, world):
    kv = _mk("dist_async", num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="dcasgd", lr=0.01))
    kv.init("w", torch.ones(4))
    for _ in range(3):
        kv.push("w", torch.full((4,), 0.5))
        out = torch.empty(4)
        kv.pull("w", out)
    assert torch.isfinite(out).all()
    # all ranks agree
    import torch.distributed as dist
    ref_out = out.clone()
    dist.broadcast(ref_out, src=0)
    assert torch.allclose(out, ref_out)


def test_dcasgd_async_ws4():
    run_dist(4, _dcasgd)


# ---------------------------------------------------------------------------
# WAN token bucket accounting
# ---------------------------------------------------------------------------

def _wan_accounting(rank, world):
    kv = _mk(num_parties=2, wan_gbps=100.0)
    n = 1000
    kv.init("w", torch.zeros(n))
    kv.push("w", torch.ones(n))
    out = torch.empty(n)
    kv.pull("w", out)
    if kv.topo.is_leader:
        assert kv.wan.total_bytes > 0
    # dense fsa sharded: push reduce (n*4*(1/2)) + pull broadcast (n*4*1)
    assert torch.allclose(out, torch.full((n,), 4.0))


def test_wan_accounting_ws4():
    run_dist(4, _wan_accounting)


# ---------------------------------------------------------------------------
# 2bit intra-party compression
# ---------------------------------------------------------------------------

def _2bit(rank, world):
    kv = _mk(num_parties=1)
    kv.set_gradient_compression({"type": "2bit", "threshold": 0.5})
    n = 100
    kv.init("w", torch.zeros(n))
    g = torch.full((n,), 0.7)  # above threshold -> each worker emits +0.5
    kv.push("w", g)
    out = torch.empty(n)
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((n,), 0.5 * world)), (rank, out)
    # residual keeps the remainder (0.2) per worker
    kv.push("w", torch.full((n,), 0.4))  # residual .2+.4=.6 -> emit .5
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((n,), 0.5 * world)), (rank, out)


def test_2bit_ws2():
    run_dist(2, _2bit)


# ---------------------------------------------------------------------------
# P3/MultiGPS big-tensor slicing across leaders
# ---------------------------------------------------------------------------

def _sliced_big_tensor(rank, world):
    kv = _mk(num_parties=2, bigarray_bound=100)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    n = 1001  # odd size: exercises padding
    kv.init("big", torch.ones(n))
    assert kv.keys["big"].sliced
    kv.push("big", torch.full((n,), 1.0))  # sum over 4 workers = 4
    out = torch.empty(n)
    kv.pull("big", out)
    assert torch.allclose(out, torch.full((n,), 0.6), atol=1e-6), (rank, out)
    # small key stays unsliced
    kv.init("small", torch.zeros(10))
    assert not kv.keys["small"].sliced
    kv.push("small", torch.ones(10))
    kv.pull("small", out[:10])
    assert torch.allclose(out[:10], torch.full((10,), -0.4), atol=1e-6)


def test_sliced_big_tensor_ws4():
    run_dist(4, _sliced_big_tensor)


# ---------------------------------------------------------------------------
# uneven party sizes (explicit GEOMX_PARTY_SIZES)
# ---------------------------------------------------------------------------

def _uneven_parties(rank, world):
    kv = _mk(num_parties=2, party_sizes=[1, 3])
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    kv.init("w", torch.ones(6))
    kv.push("w", torch.full((6,), 1.0))  # sum over all 4 workers = 4
    out = torch.empty(6)
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((6,), 0.6), atol=1e-6), (rank, out)
    # rank 0 is a whole party of 1; ranks 1..3 are party 1
    assert kv.num_workers == (1 if rank == 0 else 3)


def test_uneven_parties_ws4():
    run_dist(4, _uneven_parties)


# ---------------------------------------------------------------------------
# DGT through the kvstore WAN tier
# ---------------------------------------------------------------------------

def _dgt_kv(rank, world):
    kv = _mk(num_parties=2, enable_dgt=3, dgt_k=0.5, dgt_block_size=256)
    kv.set_gradient_compression({"type": "dgt"})
    n = 1024
    kv.init("w", torch.zeros(n))
    torch.manual_seed(5)
    g = torch.randn(n)
    kv.push("w", g)
    out = torch.empty(n)
    kv.pull("w", out)
    # lossy on unimportant chunks but all ranks identical and close to 4g
    import torch.distributed as dist
    ref = out.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(out, ref)
    rel = (out - 4 * g).abs().mean() / (4 * g).abs().mean()
    assert rel < 0.2, rel  # 4-bit chunks: bounded loss


def test_dgt_kvstore_ws4():
    run_dist(4, _dgt_kv)


# ---------------------------------------------------------------------------
# 2bit intra-party + dense inter-party (two-tier compose)
# ---------------------------------------------------------------------------

def _2bit_two_tier(rank, world):
    kv = _mk(num_parties=2)
    kv.set_gradient_compression({"type": "2bit", "threshold": 0.5})
    n = 64
    kv.init("w", torch.zeros(n))
    kv.push("w", torch.full((n,), 0.7))  # each worker quantizes to +0.5
    out = torch.empty(n)
    kv.pull("w", out)
    # party sum = 2*0.5 = 1.0; the WAN tier re-quantizes with thr 0.5:
    # emits 2 codes worth 0.5 (residual 0); inter-party sum = 2.0...
    # 2bit can only emit +-thr once per element per push, so 1.0 emits
    # 0.5 and keeps 0.5 in the leader residual; sum over parties = 1.0
    assert torch.allclose(out, torch.full((n,), 0.5 * 2)), (rank, out)
    # second push flushes the residual: leaders now emit 0.5 (residual)
    # + 0.5 (new party sum of 2x0.5... each worker: residual .2+.4=.6)
    kv.push("w", torch.full((n,), 0.4))
    kv.pull("w", out)
    # workers: residual 0.2+0.4=0.6 -> emit 0.5 each; party sum 1.0;
    # leader residual 0.5+1.0=1.5 -> emit 0.5 (single shot), keep 1.0
    assert torch.allclose(out, torch.full((n,), 0.5 * 2)), (rank, out)


def test_2bit_two_tier_ws4():
    run_dist(4, _2bit_two_tier)


# ---------------------------------------------------------------------------
# list-form API across ranks
# ---------------------------------------------------------------------------

def _list_forms(rank, world):
    kv = _mk(num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    kv.init([0, 1], [torch.ones(4), torch.zeros(6)])
    kv.push([0, 1], [torch.ones(4), torch.full((6,), 0.5)])
    o0, o1 = torch.empty(4), torch.empty(6)
    kv.pull([0, 1], [o0, o1])
    # 4 workers: w0 = 1 - 0.1*4*1, w1 = 0 - 0.1*4*0.5
    assert torch.allclose(o0, torch.full((4,), 0.6), atol=1e-6), (rank, o0)
    assert torch.allclose(o1, torch.full((6,), -0.2), atol=1e-6), (rank, o1)


def test_list_forms_ws4():
    run_dist(4, _list_forms)


# ---------------------------------------------------------------------------
# Row-sparse pull: rows-only wire (kvstore_dist.h:900 EncodeRowSparseKey)
# ---------------------------------------------------------------------------

def _row_sparse_hier(rank, world):
    kv = _mk(num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    torch.manual_seed(0)
    w = torch.randn(12, 4)
    kv.init("emb", w)
    g = torch.zeros(12, 4)
    g[rank] = 1.0  # each worker touches a different row
    kv.push("emb", g)
    # dense pull = golden model
    dense = torch.empty(12, 4)
    kv.pull("emb", dense)
    # per-worker distinct (repeated, unsorted) id lists
    ids = torch.tensor([rank, (rank + 5) % 12, rank, 11 - rank])
    out = torch.empty(len(ids), 4)
    kv.row_sparse_pull("emb", out, ids)
    assert torch.allclose(out, dense[ids], atol=1e-6), \
        (rank, out, dense[ids])


def test_row_sparse_pull_rows_only_ws4():
    run_dist(4, _row_sparse_hier)


def _row_sparse_sharded_dense(rank, world):
    # sharded dense mode (owner leader holds the value): the leader
    # tier must fetch only the requested-row union from the owner
    kv = _mk(num_parties=2)
    kv.global_mode = "sharded"
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.5))
    torch.manual_seed(1)
    w = torch.randn(10, 3)
    kv.init("emb", w)
    g = torch.ones(10, 3) * (rank + 1)
    kv.push("emb", g)
    dense = torch.empty(10, 3)
    kv.pull("emb", dense)
    ids = torch.tensor([2, 7, 2])
    out = torch.empty(3, 3)
    kv.row_sparse_pull("emb", out, ids)
    assert torch.allclose(out, dense[ids], atol=1e-6), (rank, out)


def test_row_sparse_pull_sharded_ws2():
    run_dist(2, _row_sparse_sharded_dense)


# ---------------------------------------------------------------------------
# Async push: deferred WAN tier must flush in the SAME order on every
# rank (collective matching), even when ranks pull in different orders
# ---------------------------------------------------------------------------

def _async_push_flush_order(rank, world):
    # REPLICATED global mode: pull has no leader-tier wire, so ranks may
    # pull in different orders — what must stay matched are the DEFERRED
    # push collectives, which flush in deterministic (priority, seq)
    # order at the first pull regardless of which key it names.
    kv = _mk(num_parties=2)
    kv.global_mode = "replicated"
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    torch.manual_seed(0)
    keys = ["a", "b", "c", "d", "e"]
    for k in keys:
        kv.init(k, torch.full((8,), 1.0))
    for i, k in enumerate(keys):
        kv.push(k, torch.full((8,), float(rank + 1)), priority=-i)
    assert len(kv._pending) == len(keys)
    order = keys[rank:] + keys[:rank]   # divergent pull orders
    outs = {}
    for k in order:
        o = torch.empty(8)
        kv.pull(k, o)
        outs[k] = o
    total = sum(r + 1 for r in range(world))
    for k in keys:
        expect = 1.0 - 0.1 * total
        assert torch.allclose(outs[k], torch.full((8,), expect),
                              atol=1e-5), (k, outs[k][0])


def test_async_push_flush_order_ws2():
    run_dist(2, _async_push_flush_order)


def _async_push_sharded_same_order(rank, world):
    # sharded mode keeps the SPMD contract (same pull order on every
    # rank); interleaved repushes exercise the flush-on-repush path
    kv = _mk(num_parties=2)
    kv.set_optimizer(OptimizerSpec(name="sgd", lr=0.1))
    keys = ["a", "b", "c"]
    for k in keys:
        kv.init(k, torch.full((4,), 1.0))
    for i, k in enumerate(keys):
        kv.push(k, torch.full((4,), float(rank + 1)), priority=-i)
    kv.push("a", torch.full((4,), float(rank + 1)))  # repush -> flush
    outs = {}
    for k in keys:
        o = torch.empty(4)
        kv.pull(k, o)
        outs[k] = o
    total = sum(r + 1 for r in range(world))
    assert torch.allclose(outs["a"], torch.full((4,), 1.0 - 0.2 * total),
                          atol=1e-5)
    for k in ("b", "c"):
        assert torch.allclose(outs[k], torch.full((4,), 1.0 - 0.1 * total),
                              atol=1e-5)


def test_async_push_flush_order_ws4():
    run_dist(4, _async_push_sharded_same_order)



def _bsc_dgt_kv(rank, world):
    kv = _mk(num_parties=2, dgt_block_size=256, dgt_k=0.5)
    kv.set_gradient_compression({"type": "bsc_dgt", "threshold": 0.25,
                                 "size_lower_bound": 256})
    torch.manual_seed(0)
    n = 512
    kv.init("w", torch.zeros(n))
    # structured grad so the BSC top-k keeps a well-separated set
    g = torch.zeros(n)
    g[rank::8] = 5.0
    kv.push("w", g)
    out = torch.empty(n)
    kv.pull("w", out)
    # no optimizer: pull returns the aggregated (decompressed) grad;
    # values survive the BSC select + DGT 4-bit tier within chunk error
    sel = out.abs() > 1.0
    assert sel.sum() > 0
    assert torch.isfinite(out).all()
    # every rank agrees (replayed deterministic exchange)
    import torch.distributed as dist
    ref = out.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(out, ref, atol=1e-6)


def test_bsc_dgt_kvstore_ws4():
    run_dist(4, _bsc_dgt_kv)


def _bsc_small_key_gate(rank, world):
    """Tiny keys (biases) under bsc must ship PLAIN (reference size
    gate): before r02's fix their bsc_capacity was 0 and the gradient
    silently vanished."""
    kv = _mk(num_parties=2)
    kv.set_gradient_compression({"type": "bsc", "threshold": 0.01})
    kv.init("bias", torch.zeros(16))  # capacity would be int(16*0.01)=0
    kv.push("bias", torch.ones(16))
    out = torch.empty(16)
    kv.pull("bias", out)
    assert torch.allclose(out, torch.full((16,), float(world)), atol=1e-5)


def test_bsc_small_key_plain_ws4():
    run_dist(4, _bsc_small_key_gate)


def _bsc_dgt_learns(rank, world):
    """The composed bsc_dgt kvstore path must LEARN, not just run:
    update-on-worker training on separable data drops the loss."""
    kv = _mk(num_parties=2, dgt_block_size=256, dgt_k=0.5)
    kv.set_gradient_compression({"type": "bsc_dgt", "threshold": 0.1,
                                 "size_lower_bound": 64})
    torch.manual_seed(0)
    net = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                              torch.nn.Linear(64, 4))
    params = [p for p in net.parameters()]
    for i, p in enumerate(params):
        kv.init(i, p.data)
        out = torch.empty_like(p.data)
        kv.pull(i, out)
        with torch.no_grad():
            p.copy_(out)
    opt = torch.optim.SGD(net.parameters(), lr=0.05)
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(64, 32, generator=g)
    y = (x[:, 0] > 0).long() + 2 * (x[:, 1] > 0).long()
    first = last = None
    for it in range(60):
        loss = torch.nn.functional.cross_entropy(net(x), y)
        opt.zero_grad()
        loss.backward()
        for i, p in enumerate(params):
            kv.push(i, p.grad / world, priority=-i)
        for i, p in enumerate(params):
            gagg = torch.empty_like(p.grad)
            kv.pull(i, gagg, priority=-i)
            p.grad.copy_(gagg)
        opt.step()
        if first is None:
            first = float(loss)
        last = float(loss)
    assert last < first * 0.5, (first, last)


def test_bsc_dgt_learns_ws2():
    run_dist(2, _bsc_dgt_learns)


def _2bit_learns(rank, world):
    """2bit error-feedback on BOTH tiers must learn (the residual
    carries the quantization error across steps)."""
    kv = _mk(num_parties=2)
    kv.set_gradient_compression({"type": "2bit", "threshold": 0.02})
    torch.manual_seed(0)
    net = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                              torch.nn.Linear(64, 4))
    params = list(net.parameters())
    for i, p in enumerate(params):
        kv.init(i, p.data)
        out = torch.empty_like(p.data)
        kv.pull(i, out)
        with torch.no_grad():
            p.copy_(out)
    opt = torch.optim.SGD(net.parameters(), lr=0.05)
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(64, 32, generator=g)
    y = (x[:, 0] > 0).long() + 2 * (x[:, 1] > 0).long()
    first = last = None
    for it in range(120):
        loss = torch.nn.functional.cross_entropy(net(x), y)
        opt.zero_grad()
        loss.backward()
        for i, p in enumerate(params):
            kv.push(i, p.grad / world, priority=-i)
        for i, p in enumerate(params):
            gagg = torch.empty_like(p.grad)
            kv.pull(i, gagg, priority=-i)
            p.grad.copy_(gagg)
        opt.step()
        if first is None:
            first = float(loss)
        last = float(loss)
    assert last < first * 0.6, (first, last)


def test_2bit_learns_ws4():
    run_dist(4, _2bit_learns)

"""True-async (store-transport) global tier: parties proceed at their
own pace; the global server applies every push on arrival."""

import time

import torch

from dist_helpers import run_dist

from geomx_amd import Config
from geomx_amd.kvstore import create
from geomx_amd.kvstore.optimizer import OptimizerSpec
from geomx_amd.topology import init_topology


def _mk_async(**over):
    cfg = Config.from_env(num_parties=2, backend="gloo", device="cpu",
                          mode="dist_async", async_transport="store", **over)
    topo = init_topology(2, None, "gloo", "cpu")
    return create("dist_async", cfg=cfg, topo=topo)


def _fast_party_not_stalled(rank, world):
    kv = _mk_async()
    n = 256
    kv.init("w", torch.zeros(n))
    party = kv.topo.party_id
    out = torch.empty(n)

    fast_iters, slow_iters = 10, 3
    t0 = time.perf_counter()
    if party == 0:          # fast party
        for _ in range(fast_iters):
            kv.push("w", torch.ones(n))
            kv.pull("w", out)
        fast_dt = time.perf_counter() - t0
        # slow party sleeps 0.25 s per iteration; synchronous collectives
        # would drag us to >= 0.75 s
        assert fast_dt < 0.6, fast_dt
    else:                   # slow party
        for _ in range(slow_iters):
            time.sleep(0.25)
            kv.push("w", torch.ones(n))
            kv.pull("w", out)
    kv.barrier()
    # drain: server applies every push (no optimizer -> accumulate)
    if kv._aps is not None and kv._aps.is_server:
        assert kv._aps.drain(timeout_s=30)
        assert kv._aps.applied == fast_iters + slow_iters
    kv.barrier()
    kv.pull("w", out)
    # total = 13 pushes of ones, each a party sum over 2 workers = 2.0
    assert torch.allclose(out, torch.full((n,), 2.0 * 13)), out[0]
    kv.close()


def test_async_store_no_stall_ws4():
    run_dist(4, _fast_party_not_stalled)


def _async_with_optimizer(rank, world):
    kv = _mk_async()
    kv.set_optimizer(OptimizerSpec("dcasgd", lr=0.01))
    n = 64
    kv.init("w", torch.ones(n))
    out = torch.empty(n)
    for _ in range(4):
        kv.push("w", torch.full((n,), 0.5))
        kv.pull("w", out)
    kv.barrier()
    if kv._aps is not None and kv._aps.is_server:
        assert kv._aps.drain(timeout_s=30)
        assert kv._aps.applied == 8  # 4 per party
    kv.barrier()
    kv.pull("w", out)
    assert torch.isfinite(out).all()
    assert (out != 1.0).any()  # moved from init
    kv.close()


def test_async_store_dcasgd_ws4():
    run_dist(4, _async_with_optimizer)


def _elastic_rejoin(rank, world):
    """ps-lite is_recovery parity (van.cc:372-394): a restarted leader
    resumes its push sequence; a restarted server adopts the published
    params + consumed counters — nothing is lost or double-applied."""
    import torch.distributed as dist
    from geomx_amd.kvstore.async_ps import AsyncPSGlobal

    kv = _mk_async()
    n = 32
    kv.init("w", torch.zeros(n))
    out = torch.empty(n)
    for _ in range(2):
        kv.push("w", torch.ones(n))
        kv.pull("w", out)
    kv.barrier()

    store = dist.distributed_c10d._get_default_store()
    if kv._aps is not None and kv._aps.is_server:
        assert kv._aps.drain(timeout_s=30)
        kv._aps.stop()
        # server restarts: a fresh endpoint on the same store must adopt
        # the live state, NOT the (different) init value it is given
        srv2 = AsyncPSGlobal(store, kv.topo, kv._device)
        srv2.register("w", torch.full((n,), 5.0))
        assert not torch.allclose(srv2._stored["w"], torch.full((n,), 5.0))
        assert srv2._seen[("w", 0)] == 2 and srv2._seen[("w", 1)] == 2
        srv2.start()
        kv._aps = srv2
    else:
        # party-1 leader restarts: push counter resumes where it left off
        cli2 = AsyncPSGlobal(store, kv.topo, kv._device)
        cli2.register("w", torch.zeros(n))
        assert cli2._push_seq["w"] == 2, cli2._push_seq
        kv._aps = cli2
    kv.barrier()

    kv.push("w", torch.ones(n))
    kv.pull("w", out)
    kv.barrier()
    if kv._aps.is_server:
        assert kv._aps.drain(timeout_s=30)
        assert kv._aps.applied == 2  # only the two post-restart pushes
    kv.barrier()
    kv.pull("w", out)
    # 6 total pushes of a ones party-sum, accumulate mode
    assert torch.allclose(out, torch.full((n,), 6.0)), out[0]
    kv.close()


def test_async_store_elastic_rejoin_ws2():
    run_dist(2, _elastic_rejoin)


def _stress(rank, world):
    """Race smoke (SURVEY §5.2 stance): many keys, interleaved pushes
    and pulls from both parties while the server's consumer thread
    applies concurrently — totals must come out exact."""
    kv = _mk_async()
    nkeys, iters, n = 6, 12, 17
    for k in range(nkeys):
        kv.init(k, torch.zeros(n))
    out = torch.empty(n)
    for it in range(iters):
        for k in range(nkeys):
            kv.push(k, torch.full((n,), float(k + 1)))
        # pulls are party-collective (intra-DC stays synchronous in
        # MixedSync) — the condition must be party-uniform
        party = kv.topo.party_id
        if it % 3 == party % 3:
            kv.pull((it + party) % nkeys, out)  # reads racing the consumer
    kv.barrier()
    if kv._aps is not None and kv._aps.is_server:
        assert kv._aps.drain(timeout_s=60)
        assert kv._aps.applied == 2 * nkeys * iters
    kv.barrier()
    for k in range(nkeys):
        kv.pull(k, out)
        # 2 parties x iters pushes, party sum = 2 workers x (k+1)
        expect = 2 * iters * 2.0 * (k + 1)
        assert torch.allclose(out, torch.full((n,), expect)), (k, out[0])
    kv.close()


def test_async_store_stress_ws4():
    run_dist(4, _stress)


def _lossy_transport(rank, world):
    """PS_DROP_MSG parity: 30% of push transmissions are dropped; the
    resender (src/resender.h analog) retransmits un-ACKed pushes until
    the server has applied every one — totals come out exact."""
    kv = _mk_async()
    n = 64
    kv.init("w", torch.zeros(n))
    aps = kv._aps
    if aps is not None:
        aps.drop_pct = 0.3
        aps.resend_timeout_s = 0.02
    out = torch.empty(n)
    iters = 8
    for _ in range(iters):
        kv.push("w", torch.ones(n))
        kv.pull("w", out)
    if aps is not None:
        assert aps.flush(timeout_s=30)   # resend until all ACKed
    kv.barrier()
    if aps is not None and aps.is_server:
        assert aps.drain(timeout_s=30)
        assert aps.applied == 2 * iters  # nothing lost, nothing doubled
        assert aps.lost == 0
    kv.barrier()
    kv.pull("w", out)
    assert torch.allclose(out, torch.full((n,), 2.0 * 2 * iters)), out[0]
    kv.close()


def test_async_store_lossy_transport_ws4():
    run_dist(4, _lossy_transport)


def _dead_producer_gap(rank, world):
    """A producer that dies with a push in flight leaves an
    unrepairable gap: the server skips it after skip_timeout_s, counts
    it in `lost`, and the stream continues — a lost gradient delays
    nobody (the reference's lossy-channel stance)."""
    kv = _mk_async()
    n = 16
    kv.init("w", torch.zeros(n))
    aps = kv._aps
    out = torch.empty(n)

    kv.push("w", torch.ones(n))
    kv.pull("w", out)
    kv.barrier()

    if aps is not None:
        aps.skip_timeout_s = 0.5
        if not aps.is_server:
            # party 1's leader: this push is "lost with the producer" —
            # 100% drop and then wipe the resend queue (process death)
            aps.drop_pct = 1.0
            kv.push("w", torch.ones(n))
            aps._unacked.clear()
            aps.drop_pct = 0.0
    kv.barrier()
    # a later push announces a higher seq, exposing the gap
    kv.push("w", torch.ones(n))
    kv.pull("w", out)
    if aps is not None:
        assert aps.flush(timeout_s=30)
    kv.barrier()
    if aps is not None and aps.is_server:
        assert aps.drain(timeout_s=30)
        assert aps.lost == 1, aps.lost
        # pushes: 2 parties x (1 + 1) + party1's lost one never applied
        assert aps.applied == 4, aps.applied
    kv.barrier()
    kv.pull("w", out)
    # 4 applied pushes, each a single-worker party sum of ones
    assert torch.allclose(out, torch.full((n,), 4.0)), out[0]
    kv.close()


def test_async_store_dead_producer_gap_ws2():
    run_dist(2, _dead_producer_gap)

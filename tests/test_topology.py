"""Unit tests for the HiPS topology resolution (no process group)."""

import pytest

from geomx_amd.topology import Topology, _resolve_party_sizes, init_topology


def test_resolve_uniform():
    assert _resolve_party_sizes(8, 2, None) == [4, 4]
    assert _resolve_party_sizes(8, 4, None) == [2, 2, 2, 2]


def test_resolve_explicit():
    assert _resolve_party_sizes(4, 2, [1, 3]) == [1, 3]
    with pytest.raises(ValueError):
        _resolve_party_sizes(4, 2, [1, 2])
    with pytest.raises(ValueError):
        _resolve_party_sizes(7, 2, None)


def test_single_process_topology():
    t = init_topology(1)
    assert t.world_size == 1 and t.rank == 0
    assert t.is_leader and t.is_master_worker
    assert t.num_workers == 1 and t.num_all_workers == 1
    assert t.leader_rank == 0


def test_party_of():
    t = Topology(rank=5, world_size=8, party_sizes=[2, 3, 3], party_id=2,
                 party_rank=0, party_group=None, leader_group=None,
                 party_groups=[], leader_ranks=[0, 2, 5],
                 party_ranks=[5, 6, 7], backend="gloo")
    assert t.party_of(0) == 0
    assert t.party_of(1) == 0
    assert t.party_of(2) == 1
    assert t.party_of(4) == 1
    assert t.party_of(7) == 2
    with pytest.raises(ValueError):
        t.party_of(8)
    assert t.is_leader
    assert t.leader_index == 2
    assert not t.is_master_worker


def test_config_validation_errors():
    import pytest

    from geomx_amd import Config

    with pytest.raises(ValueError):
        Config.from_env(mode="nope").validate()
    with pytest.raises(ValueError):
        Config.from_env(compression="zip").validate()
    with pytest.raises(ValueError):
        Config.from_env(bsc_ratio=1.5).validate()
    with pytest.raises(ValueError):
        Config.from_env(num_parties=0).validate()
    with pytest.raises(ValueError):
        Config.from_env(hfa_k2=0).validate()
    with pytest.raises(ValueError):
        Config.from_env(async_transport="udp").validate()


def test_kvstore_invalid_global_mode():
    import pytest

    from geomx_amd import Config
    from geomx_amd.kvstore.dist import KVStoreDist

    with pytest.raises(ValueError):
        KVStoreDist(Config.from_env(), global_mode="ring")


def test_explicit_sharded_overrides_ts():
    """An explicit global_mode wins over the ENABLE_TS default; the
    relay tier only attaches to the replicated tier."""
    from geomx_amd import Config
    from geomx_amd.kvstore import create

    kv = create("dist_sync", cfg=Config.from_env(enable_ts=True),
                global_mode="sharded")
    assert kv.global_mode == "sharded" and kv._ts is None

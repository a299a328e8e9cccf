"""Multi-node cluster executor: least-loaded spread across per-node
executors (simulated with two loopback 'hosts')."""

import asyncio
import sys

import pytest

from covalent_ssh_plugin_amd.cluster import SSHClusterExecutor, _parse_host


def test_parse_host_specs():
    assert _parse_host("alice@node0") == {"hostname": "node0", "username": "alice"}
    assert _parse_host("node1:2222") == {"hostname": "node1", "ssh_port": 2222}
    assert _parse_host("bare") == {"hostname": "bare"}
    assert _parse_host({"hostname": "h", "gpu_slots": 4}) == {
        "hostname": "h",
        "gpu_slots": 4,
    }
    with pytest.raises(ValueError):
        _parse_host("bad@spec@oops")


def _cluster(tmp_path, **kw):
    homes = []
    hosts = []
    for i in range(2):
        home = tmp_path / f"host{i}"
        home.mkdir()
        homes.append(home)
        hosts.append({"local_home": str(home)})
    defaults = dict(
        transport="local",
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
    )
    defaults.update(kw)
    return SSHClusterExecutor(hosts, **defaults), homes


def test_cluster_roundtrip_and_spread(tmp_path):
    cluster, homes = _cluster(tmp_path, persistent_workers=True, cpu_workers=1)

    def whereami(i):
        import os

        return i, os.getcwd()

    async def main():
        return await asyncio.gather(
            *[
                cluster.execute(whereami, [i], {}, dispatch_id="cl", node_id=i)
                for i in range(8)
            ]
        )

    out = asyncio.run(main())
    assert [i for i, _ in out] == list(range(8))
    # work landed on BOTH simulated hosts
    used_homes = {cwd.split("/covalent-workdir")[0] for _, cwd in out}
    assert len(used_homes) == 2, used_homes


def test_cluster_empty_hosts_rejected():
    with pytest.raises(ValueError):
        SSHClusterExecutor([])


def test_cluster_stats_and_capacity(tmp_path):
    cluster, _ = _cluster(tmp_path, gpu_slots=8)
    assert cluster.capacity == 16
    asyncio.run(cluster.execute(lambda: 1, [], {}))
    stats = cluster.stats()
    assert len(stats) == 2
    assert sum(s["counters"]["tasks"] for s in stats.values()) == 1


def test_cluster_prewarm(tmp_path):
    cluster, _ = _cluster(tmp_path, persistent_workers=True, cpu_workers=2)
    n = asyncio.run(cluster.prewarm())
    assert n == 4  # 2 hosts x 2 cpu workers
    assert asyncio.run(cluster.execute(lambda: "warm", [], {})) == "warm"

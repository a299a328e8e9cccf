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


def _mixed_cluster(tmp_path, **kw):
    """One healthy loopback node + one node whose home doesn't exist
    (connect fails)."""
    good = tmp_path / "good"
    good.mkdir()
    hosts = [
        {"local_home": str(tmp_path / "missing"), "hostname": "deadnode"},
        {"local_home": str(good), "hostname": "goodnode"},
    ]
    defaults = dict(
        transport="local",
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
        retry_connect=False,
    )
    defaults.update(kw)
    return SSHClusterExecutor(hosts, **defaults)


def test_failover_to_healthy_node(tmp_path):
    """A node failing BEFORE execution (connect error) is cooled down
    and the electron transparently lands on the healthy node."""
    cluster = _mixed_cluster(tmp_path)

    def add(x, y):
        return x + y

    async def main():
        return await asyncio.gather(
            *[
                cluster.execute(add, [i, i], {}, dispatch_id="fo", node_id=i)
                for i in range(6)
            ]
        )

    assert asyncio.run(main()) == [0, 2, 4, 6, 8, 10]
    # the dead node is marked unhealthy (cooldown active)
    import time

    assert cluster._unhealthy_until[0] > time.monotonic()


def test_failover_disabled_raises(tmp_path):
    from covalent_ssh_plugin_amd.ssh import SSHConnectError

    cluster = _mixed_cluster(tmp_path, failover=False)

    def add(x, y):
        return x + y

    async def main():
        # dispatch until the dead node is picked (least-loaded +
        # round-robin: sequential singles alternate nodes)
        for i in range(2):
            await cluster.execute(add, [1, 1], {}, dispatch_id="nf", node_id=i)

    with pytest.raises(SSHConnectError):
        asyncio.run(main())


def test_all_nodes_down_raises(tmp_path):
    from covalent_ssh_plugin_amd.ssh import SSHConnectError

    hosts = [
        {"local_home": str(tmp_path / "m1"), "hostname": "dead1"},
        {"local_home": str(tmp_path / "m2"), "hostname": "dead2"},
    ]
    cluster = SSHClusterExecutor(
        hosts,
        transport="local",
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
        retry_connect=False,
    )

    def add(x, y):
        return x + y

    with pytest.raises(SSHConnectError):
        asyncio.run(cluster.execute(add, [1, 2], {}))


def test_task_errors_do_not_fail_over(tmp_path):
    """A task exception (post-execution) must surface, never retry on
    another node — the task already ran once."""
    exec_log = tmp_path / "execs"

    def boom():
        with open(str(exec_log), "a") as f:
            f.write("ran\n")
        raise ValueError("task bug")

    cluster, _ = _cluster(tmp_path)
    with pytest.raises(ValueError, match="task bug"):
        asyncio.run(cluster.execute(boom, [], {}))
    assert exec_log.read_text().count("ran") == 1

"""GPU slot scheduler tests (no GPU needed — pure accounting)."""

import asyncio

from covalent_ssh_plugin_amd.gpu.slots import SlotTable, get_slot_table


def test_round_robin_spread():
    async def main():
        table = SlotTable(num_gpus=4)
        held = [await table.acquire() for _ in range(4)]
        assert sorted(s.gpu_id for s in held) == [0, 1, 2, 3]
        for s in held:
            await s.release()

    asyncio.run(main())


def test_least_recently_used_preference():
    async def main():
        table = SlotTable(num_gpus=4)
        s0 = await table.acquire()
        await s0.release()
        # next acquire should prefer a *different* gpu (round-robin), not
        # immediately reuse gpu 0
        s1 = await table.acquire()
        assert s1.gpu_id != s0.gpu_id
        await s1.release()

    asyncio.run(main())


def test_blocking_and_release():
    async def main():
        table = SlotTable(num_gpus=1)
        s = await table.acquire()
        assert table.in_use == 1

        grabbed = asyncio.Event()

        async def waiter():
            s2 = await table.acquire()
            grabbed.set()
            await s2.release()

        task = asyncio.create_task(waiter())
        await asyncio.sleep(0.05)
        assert not grabbed.is_set()  # blocked while slot held
        await s.release()
        await asyncio.wait_for(task, timeout=2)
        assert grabbed.is_set()

    asyncio.run(main())


def test_context_manager_releases_on_error():
    async def main():
        table = SlotTable(num_gpus=1)
        try:
            async with await table.acquire():
                raise RuntimeError("task failed")
        except RuntimeError:
            pass
        assert table.in_use == 0
        # slot is reusable afterwards
        s = await table.acquire()
        await s.release()

    asyncio.run(main())


def test_env_injection():
    async def main():
        table = SlotTable(num_gpus=8)
        s = await table.acquire()
        env = s.env()
        assert env["CSP_GPU_SLOT"] == str(s.gpu_id)
        await s.release()

    asyncio.run(main())


def test_oversubscription():
    async def main():
        table = SlotTable(num_gpus=2, slots_per_gpu=2)
        held = [await table.acquire() for _ in range(4)]
        per_gpu = {}
        for s in held:
            per_gpu[s.gpu_id] = per_gpu.get(s.gpu_id, 0) + 1
        assert per_gpu == {0: 2, 1: 2}
        for s in held:
            await s.release()

    asyncio.run(main())


def test_table_shared_across_instances():
    """Two executors pointing at the same endpoint must share accounting
    (SURVEY.md §7 'Hard parts')."""
    t1 = get_slot_table(("ssh", "h", "u"), num_gpus=8)
    t2 = get_slot_table(("ssh", "h", "u"), num_gpus=8)
    assert t1 is t2
    t3 = get_slot_table(("ssh", "other", "u"), num_gpus=8)
    assert t3 is not t1


def test_fair_distribution_under_concurrency():
    async def main():
        table = SlotTable(num_gpus=8)
        counts = [0] * 8

        async def job(i):
            async with await table.acquire() as s:
                counts[s.gpu_id] += 1
                await asyncio.sleep(0.001)

        await asyncio.gather(*[job(i) for i in range(64)])
        assert sum(counts) == 64
        # round-robin should keep the spread tight
        assert max(counts) - min(counts) <= 2, counts

    asyncio.run(main())

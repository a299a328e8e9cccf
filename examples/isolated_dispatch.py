#!/usr/bin/env python3
"""Fork-isolated dispatch: fresh-process-per-electron semantics (like
the classic spawn-per-task stub) at fork cost — measured 2.67 ms vs
36.6 ms per no-op electron on an MI355X box.

Each electron runs in its own forked child of a warm zygote: no module
state, environment mutation or HIP context survives into the next
electron, and a hard crash (segfault, os._exit) is contained and
reported as a clean task error without losing the worker.
"""

import asyncio
import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from covalent_ssh_plugin_amd import SSHExecutor


def leaky_electron(i):
    import os

    os.environ["LEAK"] = f"from-electron-{i}"  # dies with the child
    return (i, os.getpid(), os.environ.get("LEAK"))


def check_clean():
    import os

    return os.environ.get("LEAK", "no leakage — fresh process")


def crashy():
    import os

    os._exit(42)  # simulated hard crash


async def main():
    with tempfile.TemporaryDirectory() as home, tempfile.TemporaryDirectory() as cache:
        ex = SSHExecutor(
            transport="local",
            local_home=home,
            cache_dir=cache,
            python_path=sys.executable,
            isolate_tasks=True,
            isolate_preload="none",  # cheap forks; "torch" pre-binds torch
            cpu_workers=2,
        )
        for i in range(3):
            print("electron:", await ex.execute(leaky_electron, [i], {}, node_id=i))
        print("next electron sees:", await ex.execute(check_clean, [], {}, node_id=10))
        try:
            await ex.execute(crashy, [], {}, node_id=20)
        except RuntimeError as e:
            print("crash contained:", e)
        print("still serving:", await ex.execute(check_clean, [], {}, node_id=21))
        await SSHExecutor.close_pool()


if __name__ == "__main__":
    asyncio.run(main())

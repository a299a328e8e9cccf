#!/usr/bin/env python3
"""Classic covalent usage: the plugin registers as executor="ssh" via the
covalent.executor.executor_plugins entry point (pip install -e . first,
covalent server running)."""

import covalent as ct


@ct.electron(executor="ssh")
def train(n):
    import torch

    x = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
    return (x @ x).float().mean().item()


@ct.electron
def report(value):
    return f"mean={value:.4f}"


@ct.lattice
def workflow(n):
    return report(train(n))


if __name__ == "__main__":
    dispatch_id = ct.dispatch(workflow)(4096)
    print(ct.get_result(dispatch_id, wait=True).result)

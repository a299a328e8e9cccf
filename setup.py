"""Packaging for the MI355X-native Covalent SSH executor plugin.

Registers the executor under the ``covalent.executor.executor_plugins``
entry-point group (reference setup.py:36,74-76) so ``executor="ssh"``
and ``from covalent.executor import SSHExecutor`` resolve to this plugin
when covalent is installed.  The CDNA4 HIP library is built in-tree by
``python -m covalent_ssh_plugin_amd.ops.build`` (hipcc, gfx950) and
shipped as package data when present.
"""

from pathlib import Path

from setuptools import find_packages, setup

here = Path(__file__).parent
version = (here / "VERSION").read_text().strip()

setup(
    name="covalent-ssh-plugin-amd",
    version=version,
    description=(
        "MI355X-native Covalent SSH executor: pooled multiplexed SSH dispatch "
        "onto an 8xMI355X node with per-task GPU slots, a CDNA4 warm-up/probe "
        "kernel and hipHostMalloc-pinned result staging"
    ),
    long_description=(here / "README.md").read_text(),
    long_description_content_type="text/markdown",
    packages=find_packages(include=["covalent_ssh_plugin_amd*"]),
    package_data={
        "covalent_ssh_plugin_amd": ["ops/*.so", "ops/hip/*.hip", "remote/stub_template.py"],
    },
    # runtime deps live in requirements.txt (reference setup.py:28-34
    # parses the same file; license CI checks it)
    install_requires=[
        line.strip()
        for line in (here / "requirements.txt").read_text().splitlines()
        if line.strip() and not line.startswith("#")
    ],
    extras_require={"covalent": ["covalent>=0.202.0,<1"]},
    entry_points={
        "covalent.executor.executor_plugins": [
            "ssh = covalent_ssh_plugin_amd.ssh",
        ],
    },
    python_requires=">=3.9",
    classifiers=[
        "Programming Language :: Python :: 3.9",
        "Programming Language :: Python :: 3.10",
        "Environment :: GPU",
        "Operating System :: POSIX :: Linux",
    ],
)

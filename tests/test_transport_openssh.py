"""OpenSSH ControlMaster transport unit tests (no network: the ssh
subprocess layer is faked)."""

import asyncio

import pytest

from covalent_ssh_plugin_amd.transport import TransportConnectError
from covalent_ssh_plugin_amd.transport.base import CompletedCommand
from covalent_ssh_plugin_amd.transport.openssh import OpenSSHTransport


def make_transport(tmp_path, **kw):
    kw.setdefault("hostname", "node0")
    kw.setdefault("username", "alice")
    kw.setdefault("ssh_key_file", "/keys/id")
    kw.setdefault("control_dir", str(tmp_path))
    return OpenSSHTransport(**kw)


def test_base_args_multiplexing(tmp_path):
    t = make_transport(tmp_path, port=2222)
    args = t._base_args()
    joined = " ".join(args)
    assert "ControlMaster=auto" in joined
    assert f"ControlPath={t._control_path}" in joined
    assert "ControlPersist=" in joined
    # reference host-key policy parity (known_hosts=None, ssh.py:267)
    assert "StrictHostKeyChecking=no" in joined
    assert "UserKnownHostsFile=/dev/null" in joined
    assert "BatchMode=yes" in joined
    assert ["-p", "2222"] == args[args.index("-p"):args.index("-p") + 2]
    assert ["-i", "/keys/id"] == args[args.index("-i"):args.index("-i") + 2]
    assert ["-l", "alice"] == args[args.index("-l"):args.index("-l") + 2]


def test_control_path_keyed_per_endpoint(tmp_path):
    a = make_transport(tmp_path)
    b = make_transport(tmp_path)
    c = make_transport(tmp_path, hostname="node1")
    assert a._control_path == b._control_path
    assert a._control_path != c._control_path


class FakeExec:
    """Patches _ssh_exec; records commands, returns scripted results."""

    def __init__(self, transport, results):
        self.commands = []
        self.inputs = []
        self.results = list(results)

        async def fake(command, input_data=None, timeout=None):
            self.commands.append(command)
            self.inputs.append(input_data)
            return self.results.pop(0) if self.results else CompletedCommand(0, b"", b"")

        transport._ssh_exec = fake


def test_connect_success_and_idempotence(tmp_path):
    t = make_transport(tmp_path)
    fake = FakeExec(t, [CompletedCommand(0, b"", b"")])

    asyncio.run(t.connect())
    assert t.is_connected
    assert fake.commands == ["true"]


def test_connect_failure_raises(tmp_path):
    t = make_transport(tmp_path)
    FakeExec(t, [CompletedCommand(255, b"", b"Connection refused")])
    with pytest.raises(TransportConnectError, match="refused"):
        asyncio.run(t.connect())
    assert not t.is_connected


def test_run_passes_remote_exit_through(tmp_path):
    """A remote command's nonzero exit is NOT a transport error
    (reference semantics: exit_status surfaces to policy code)."""
    t = make_transport(tmp_path)
    FakeExec(t, [CompletedCommand(0, b"", b""), CompletedCommand(3, b"", b"oops")])

    async def main():
        await t.connect()
        return await t.run("exit 3")

    proc = asyncio.run(main())
    assert proc.returncode == 3
    assert "oops" in proc.text_err()


def test_run_detects_dropped_channel(tmp_path):
    """ssh client exit 255 = the channel died -> TransportConnectError and
    the transport marks itself disconnected (pool will reconnect)."""
    t = make_transport(tmp_path)
    FakeExec(t, [CompletedCommand(0, b"", b""), CompletedCommand(255, b"", b"broken pipe")])

    async def main():
        await t.connect()
        await t.run("true")

    with pytest.raises(TransportConnectError):
        asyncio.run(main())
    assert not t.is_connected


def test_put_files_streams_tar(tmp_path):
    src = tmp_path / "payload.bin"
    src.write_bytes(b"DATA")
    t = make_transport(tmp_path)
    fake = FakeExec(t, [CompletedCommand(0, b"", b""), CompletedCommand(0, b"", b"")])

    async def main():
        await t.connect()
        await t.put_files([(str(src), ".cache/covalent/payload.bin")])

    asyncio.run(main())
    assert fake.commands[1] == "tar -xf -"
    assert fake.inputs[1] is not None and fake.inputs[1][:2] != b""
    # relative remote path -> extraction relative to remote $HOME
    import io
    import tarfile

    with tarfile.open(fileobj=io.BytesIO(fake.inputs[1])) as tf:
        assert tf.getnames() == [".cache/covalent/payload.bin"]
        assert tf.extractfile(".cache/covalent/payload.bin").read() == b"DATA"


def test_env_prefix_quoting(tmp_path):
    t = make_transport(tmp_path)
    prefix = t._env_prefix({"CSP_GPU_SLOT": "3", "X": "a b"})
    assert prefix == "export CSP_GPU_SLOT=3 X='a b' && "


def test_split_stream_sentinels():
    from covalent_ssh_plugin_amd.ssh import FusedStreamParser

    s_result, s_meta = b"\n--R--\n", b"\n--M--\n"
    task_out = b"user output \x00\xff --R not quite"
    result = b"\x80\x04binary pickle"
    meta = b'{"phases_ms": {}}'

    def parse(stream, chunk=5):
        sink = bytearray()
        p = FusedStreamParser(s_result, s_meta, sink.extend)
        for i in range(0, len(stream), chunk):
            p.feed(stream[i : i + chunk])
        p.finish()
        return (
            bytes(p.task_out),
            bytes(sink) if p.have_result else None,
            p.meta_bytes,
        )

    stream = task_out + s_result + result + s_meta + meta
    assert parse(stream) == (task_out, result, meta)

    # no sentinel at all (task failed before writing the result)
    assert parse(b"just logs") == (b"just logs", None, None)

    # result but no meta
    t, r, m = parse(task_out + s_result + result)
    assert r == result and m is None


def test_real_ssh_client_connect_refused(tmp_path):
    """Runs the REAL ssh binary against a closed local port: the
    transport must surface TransportConnectError (feeds the executor's
    retry loop), not hang or crash."""
    t = OpenSSHTransport(
        hostname="127.0.0.1",
        username="nobody",
        port=1,  # reserved port, nothing listens
        control_dir=str(tmp_path),
        connect_timeout=5,
    )
    with pytest.raises(TransportConnectError):
        asyncio.run(t.connect())
    assert not t.is_connected


def test_open_channel_argv(tmp_path, monkeypatch):
    """Worker channels ride the same multiplexed ssh argv, with the slot
    env exported in the remote command."""
    captured = {}

    async def fake_exec(*argv, **kw):
        captured["argv"] = argv

        class P:
            returncode = None
            stdin = stdout = None

        return P()

    monkeypatch.setattr(asyncio, "create_subprocess_exec", fake_exec)
    t = make_transport(tmp_path)

    async def main():
        await t.open_channel("python worker.py", env={"CSP_GPU_SLOT": "5"})

    asyncio.run(main())
    argv = captured["argv"]
    assert argv[0] == "ssh"
    assert "node0" in argv
    remote_cmd = argv[-1]
    assert remote_cmd == "export CSP_GPU_SLOT=5 && python worker.py"
    assert f"ControlPath={t._control_path}" in " ".join(argv)

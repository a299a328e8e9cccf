"""LocalTransport + tar-stream staging behavior."""

import asyncio

import pytest

from covalent_ssh_plugin_amd.transport import (
    LocalTransport,
    TransportConnectError,
    make_tar_stream,
)


def test_run_and_env(tmp_path):
    async def main():
        t = LocalTransport(home=str(tmp_path))
        await t.connect()
        proc = await t.run("pwd && echo -n $CSP_TEST_VAR", env={"CSP_TEST_VAR": "v1"})
        assert proc.ok
        lines = proc.text_out().splitlines()
        assert lines[0] == str(tmp_path)
        assert lines[1] == "v1"
        await t.close()

    asyncio.run(main())


def test_run_requires_connect(tmp_path):
    async def main():
        t = LocalTransport(home=str(tmp_path))
        with pytest.raises(TransportConnectError):
            await t.run("true")

    asyncio.run(main())


def test_stdin_streaming(tmp_path):
    async def main():
        t = LocalTransport(home=str(tmp_path))
        await t.connect()
        proc = await t.run("cat > received.bin", input_data=b"\x00\x01binary\xff")
        assert proc.ok
        assert (tmp_path / "received.bin").read_bytes() == b"\x00\x01binary\xff"

    asyncio.run(main())


def test_put_get_roundtrip(tmp_path):
    src = tmp_path / "src.txt"
    src.write_text("payload")
    home = tmp_path / "home"
    home.mkdir()

    async def main():
        t = LocalTransport(home=str(home))
        await t.connect()
        await t.put_files([(str(src), "sub/dir/dst.txt")])
        assert (home / "sub/dir/dst.txt").read_text() == "payload"
        out = tmp_path / "back.txt"
        await t.get_file("sub/dir/dst.txt", str(out))
        assert out.read_text() == "payload"

    asyncio.run(main())


def test_nonzero_exit_reported(tmp_path):
    async def main():
        t = LocalTransport(home=str(tmp_path))
        await t.connect()
        proc = await t.run("echo oops >&2; exit 3")
        assert proc.returncode == 3
        assert "oops" in proc.text_err()

    asyncio.run(main())


def test_tar_stream_relative(tmp_path):
    f = tmp_path / "a.bin"
    f.write_bytes(b"abc")
    data, base = make_tar_stream([(str(f), ".cache/covalent/a.bin")])
    assert base == ""  # extract relative to remote home
    assert data[:5] != b""


def test_tar_stream_absolute(tmp_path):
    f = tmp_path / "a.bin"
    f.write_bytes(b"abc")
    data, base = make_tar_stream([(str(f), "/opt/stage/a.bin")])
    assert base == "/"


def test_tar_stream_mixed_rejected(tmp_path):
    f = tmp_path / "a.bin"
    f.write_bytes(b"abc")
    with pytest.raises(ValueError):
        make_tar_stream([(str(f), "/abs/a.bin"), (str(f), "rel/a.bin")])


def test_put_files_empty_batch(tmp_path):
    async def main():
        t = LocalTransport(home=str(tmp_path))
        await t.connect()
        await t.put_files([])  # no-op, must not raise

    asyncio.run(main())


def test_summarize_empty_records():
    from covalent_ssh_plugin_amd.utils.timing import summarize

    assert summarize([]) == {"count": 0}


def test_pool_reconnects_dropped_transport(tmp_path):
    from covalent_ssh_plugin_amd.transport import pool as transport_pool

    async def main():
        home = tmp_path / "h"
        home.mkdir()
        made = []

        def factory():
            t = LocalTransport(home=str(home))
            made.append(t)
            return t

        t1 = await transport_pool.get_transport(("k",), factory)
        assert t1.is_connected
        await t1.close()  # connection dropped
        t2 = await transport_pool.get_transport(("k",), factory)
        assert t2 is t1  # same pooled instance...
        assert t2.is_connected  # ...reconnected
        assert len(made) == 1

    asyncio.run(main())

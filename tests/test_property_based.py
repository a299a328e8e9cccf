"""Property-based tests (hypothesis) for the protocol-critical pieces:
sentinel stream splitting, tar staging round trips, frame handling and
slot-table invariants."""

import asyncio
import io
import tarfile

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from covalent_ssh_plugin_amd.gpu.slots import SlotTable
from covalent_ssh_plugin_amd.ssh import SSHExecutor
from covalent_ssh_plugin_amd.transport.base import make_tar_stream

S_RESULT = b"\n--CSP-RESULT-feedfacefeedface--\n"
S_META = b"\n--CSP-META-feedfacefeedface--\n"

# arbitrary binary that never contains the sentinels (uuid-tokenized in
# production; the splitter contract assumes non-occurrence)
binary = st.binary(max_size=2048).filter(
    lambda b: S_RESULT not in b and S_META not in b
)


def _parse_chunked(stream: bytes, chunks):
    """Feed ``stream`` through FusedStreamParser at the given chunk
    boundaries; returns (task_out, result_bytes_or_None, meta_or_None)."""
    from covalent_ssh_plugin_amd.ssh import FusedStreamParser

    sink = bytearray()
    parser = FusedStreamParser(S_RESULT, S_META, sink.extend)
    pos = 0
    for size in chunks:
        parser.feed(stream[pos : pos + size])
        pos += size
    parser.feed(stream[pos:])
    parser.finish()
    result = bytes(sink) if parser.have_result else None
    if parser.have_result:
        import hashlib

        assert parser.sha_hex == hashlib.sha256(sink).hexdigest()
    return bytes(parser.task_out), result, parser.meta_bytes


@settings(max_examples=200, deadline=None)
@given(
    task_out=binary,
    result=binary,
    meta=binary,
    chunks=st.lists(st.integers(min_value=1, max_value=97), max_size=40),
)
def test_split_stream_roundtrip(task_out, result, meta, chunks):
    """The streaming splitter must reproduce the exact segments under
    ARBITRARY chunk boundaries (sentinels straddling reads included)."""
    stream = task_out + S_RESULT + result + S_META + meta
    t, r, m = _parse_chunked(stream, chunks)
    assert (t, r, m) == (task_out, result, meta)


@settings(max_examples=100, deadline=None)
@given(
    task_out=binary,
    chunks=st.lists(st.integers(min_value=1, max_value=97), max_size=40),
)
def test_split_stream_no_result(task_out, chunks):
    t, r, m = _parse_chunked(task_out, chunks)
    assert t == task_out and r is None and m is None


name_chars = st.text(
    alphabet=st.characters(whitelist_categories=("Ll", "Lu", "Nd"), whitelist_characters="-_."),
    min_size=1,
    max_size=24,
).filter(lambda s: s not in (".", ".."))


@settings(max_examples=50, deadline=None)
@given(
    files=st.lists(
        st.tuples(name_chars, st.binary(max_size=4096)), min_size=1, max_size=5,
        unique_by=lambda t: t[0],
    ),
    absolute=st.booleans(),
)
def test_tar_stream_roundtrip(tmp_path_factory, files, absolute):
    tmp_path = tmp_path_factory.mktemp("tar")
    pairs = []
    for i, (name, data) in enumerate(files):
        local = tmp_path / f"src{i}.bin"
        local.write_bytes(data)
        remote = (f"/stage/{name}-{i}" if absolute else f".cache/{name}-{i}")
        pairs.append((str(local), remote))
    blob, base = make_tar_stream(pairs)
    assert base == ("/" if absolute else "")
    with tarfile.open(fileobj=io.BytesIO(blob)) as tf:
        for (local, remote), (name, data) in zip(pairs, files):
            member = remote.lstrip("/") if absolute else remote
            assert tf.extractfile(member).read() == data


@settings(max_examples=50, deadline=None)
@given(
    num_gpus=st.integers(min_value=1, max_value=8),
    ops=st.lists(st.integers(min_value=0, max_value=1), min_size=1, max_size=60),
)
def test_slot_table_invariants(num_gpus, ops):
    """Random acquire/release interleavings never over-allocate a GPU and
    always restore full capacity."""

    async def main():
        table = SlotTable(num_gpus=num_gpus)
        held = []
        for op in ops:
            if op == 0 and table.in_use < table.capacity:
                held.append(await table.acquire())
            elif held:
                await held.pop().release()
            # invariant: every gpu id valid, no over-allocation
            in_use_ids = [s.gpu_id for s in held]
            assert all(0 <= g < num_gpus for g in in_use_ids)
            for g in set(in_use_ids):
                assert in_use_ids.count(g) <= table.slots_per_gpu
            assert table.in_use == len(held)
        for s in held:
            await s.release()
        assert table.in_use == 0
        # full capacity restored
        final = [await table.acquire() for _ in range(table.capacity)]
        assert sorted(s.gpu_id for s in final) == sorted(
            list(range(num_gpus)) * table.slots_per_gpu
        )
        for s in final:
            await s.release()

    asyncio.run(main())


@settings(max_examples=30, deadline=None)
@given(payloads=st.lists(st.binary(min_size=1, max_size=1 << 16), min_size=1, max_size=8))
def test_channel_frame_roundtrip(tmp_path_factory, payloads):
    """Arbitrary binary frames survive the worker framing protocol
    (cat-style echo worker).  Zero-length frames are excluded: they are
    the protocol's shutdown sentinel (production requests are pickled
    dicts, never empty)."""
    echo = (
        "import os, struct, sys\n"
        "proto = os.dup(1); os.dup2(2, 1)\n"
        "def rf():\n"
        "    h = b''\n"
        "    while len(h) < 8:\n"
        "        c = os.read(0, 8 - len(h))\n"
        "        if not c: return None\n"
        "        h += c\n"
        "    (n,) = struct.unpack('>Q', h)\n"
        "    data = b''\n"
        "    while len(data) < n:\n"
        "        c = os.read(0, n - len(data))\n"
        "        if not c: return None\n"
        "        data += c\n"
        "    return data\n"
        "while True:\n"
        "    d = rf()\n"
        "    if d is None or d == b'': break\n"
        "    os.write(proto, struct.pack('>Q', len(d))); os.write(proto, d)\n"
    )

    async def main():
        import sys

        from covalent_ssh_plugin_amd.transport.channel import open_subprocess_channel

        ch = await open_subprocess_channel([sys.executable, "-c", echo], "echo")
        try:
            for p in payloads:
                got = await ch.request(p, timeout=30)
                assert got == p
        finally:
            await ch.close()

    asyncio.run(main())


@settings(max_examples=60, deadline=None)
@given(
    num_gpus=st.integers(min_value=1, max_value=4),
    slots_per_gpu=st.integers(min_value=1, max_value=4),
    ops=st.lists(st.integers(min_value=0, max_value=1), min_size=1, max_size=80),
)
def test_subslot_invariants(num_gpus, slots_per_gpu, ops):
    """Oversubscription sub-slots: a (gpu, sub) pair is never held by
    two electrons at once, sub ids stay in range, and worker tags are
    unique across held slots (one worker process per sub-slot)."""

    async def main():
        table = SlotTable(num_gpus=num_gpus, slots_per_gpu=slots_per_gpu)
        held = []
        for op in ops:
            if op == 0 and table.in_use < table.capacity:
                held.append(await table.acquire())
            elif held:
                await held.pop().release()
            pairs = [(s.gpu_id, s.sub_id) for s in held]
            assert len(pairs) == len(set(pairs))
            assert all(0 <= sub < slots_per_gpu for _, sub in pairs)
            tags = [s.worker_tag for s in held]
            assert len(tags) == len(set(tags))
        for s in held:
            await s.release()
        assert table.in_use == 0
        # full capacity restored with every sub-slot exactly once
        final = [await table.acquire() for _ in range(table.capacity)]
        assert sorted((s.gpu_id, s.sub_id) for s in final) == sorted(
            (g, sub) for g in range(num_gpus) for sub in range(slots_per_gpu)
        )
        for s in final:
            await s.release()

    asyncio.run(main())

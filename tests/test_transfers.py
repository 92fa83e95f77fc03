"""Transfer manager: pacing, staging-rename uploads, path jail, browse +
the HTTP endpoints."""

import asyncio
import os
import time

import pytest

from selkies_amd.transfers import TransferManager, TransferPacer


@pytest.fixture()
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def test_pacer_rate(loop):
    async def main():
        pacer = TransferPacer(rate_bytes_per_s=1_000_000, burst_bytes=100_000)
        t0 = time.monotonic()
        # 600KB at 1MB/s with 100KB burst -> >= ~0.4s
        for _ in range(6):
            await pacer.consume(100_000)
        return time.monotonic() - t0

    dt = loop.run_until_complete(main())
    assert dt > 0.35, f"pacer too permissive: {dt:.2f}s"


def test_upload_staging_and_atomic_rename(tmp_path, loop):
    async def main():
        tm = TransferManager(str(tmp_path))
        saw_part = {}

        async def chunks():
            yield b"a" * 1000
            saw_part["present"] = any(
                n.startswith(".") and n.endswith(".part")
                for n in os.listdir(tmp_path))
            yield b"b" * 500

        res = await tm.upload("out.bin", chunks())
        assert res["bytes"] == 1500
        assert saw_part["present"], "upload must stage to a hidden .part"
        assert (tmp_path / "out.bin").read_bytes() == b"a" * 1000 + b"b" * 500
        assert not any(n.endswith(".part") for n in os.listdir(tmp_path))

    loop.run_until_complete(main())


def test_path_jail(tmp_path, loop):
    tm = TransferManager(str(tmp_path))
    with pytest.raises(PermissionError):
        tm.resolve("../evil")
    with pytest.raises(PermissionError):
        tm.resolve("a/../../evil")
    # absolute paths are reinterpreted relative to the jail root
    assert tm.resolve("/etc/passwd").startswith(str(tmp_path))
    assert tm.resolve("sub/ok.txt").startswith(str(tmp_path))


def test_listdir_hides_parts(tmp_path):
    (tmp_path / "visible.txt").write_text("x")
    (tmp_path / ".h.part").write_text("y")
    tm = TransferManager(str(tmp_path))
    names = [e["name"] for e in tm.listdir("")]
    assert names == ["visible.txt"]


def test_http_endpoints(tmp_path, loop):
    async def main():
        import aiohttp
        from aiohttp import web
        from selkies_amd.settings import load_settings
        from selkies_amd.stream_server import CentralizedStreamServer

        settings = load_settings(argv=[], env={
            "SELKIES_UPLOAD_DIR": str(tmp_path),
            "SELKIES_ENABLE_AUDIO": "false",
        })
        server = CentralizedStreamServer(settings)
        runner = web.AppRunner(server.app)
        await runner.setup()
        site = web.TCPSite(runner, "127.0.0.1", 0)
        await site.start()
        port = site._server.sockets[0].getsockname()[1]
        try:
            async with aiohttp.ClientSession() as sess:
                url = f"http://127.0.0.1:{port}"
                r = await sess.post(f"{url}/api/upload?name=f.txt",
                                    data=b"hello world")
                assert r.status == 200
                assert (tmp_path / "f.txt").read_text() == "hello world"
                r = await sess.get(f"{url}/api/files")
                files = await r.json()
                assert files[0]["name"] == "f.txt"
                r = await sess.get(f"{url}/api/download?name=f.txt")
                assert await r.read() == b"hello world"
                r = await sess.get(f"{url}/api/download?name=../../etc/hosts")
                assert r.status in (403, 404)
        finally:
            await runner.cleanup()

    loop.run_until_complete(main())


def test_uplink_allowance_aimd():
    """RTT inflation while transferring backs the rate off multiplicatively;
    healthy intervals creep it back up; floor/cap clamped; video share
    bounds the budget."""
    from selkies_amd.transfers import TransferPacer, UplinkAllowance
    pacer = TransferPacer(rate_bytes_per_s=10_000_000)
    a = UplinkAllowance(pacer)
    # healthy: establishes the RTT floor and creeps up
    r1 = a.observe(video_bps=0, rtt_ms=20)
    r2 = a.observe(video_bps=0, rtt_ms=21)
    assert r2 >= r1
    # congestion: rtt >> floor -> multiplicative backoff, pacer follows
    r3 = a.observe(video_bps=0, rtt_ms=200)
    assert r3 <= r2 * 0.6
    assert pacer.rate == r3
    # repeated congestion floors out
    for _ in range(30):
        r = a.observe(video_bps=0, rtt_ms=500)
    assert r == a.floor_rate
    # recovery is gradual (additive)
    r4 = a.observe(video_bps=0, rtt_ms=20)
    assert r4 <= a.floor_rate + a.CREEP_BYTES
    # a heavy video stream caps the transfer share
    a2 = UplinkAllowance(TransferPacer(rate_bytes_per_s=50_000_000))
    r5 = a2.observe(video_bps=8_000_000, rtt_ms=20)
    assert r5 <= (8_000_000 + 50_000_000 + a2.CREEP_BYTES) * 0.35 + 1

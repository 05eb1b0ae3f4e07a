"""cbresolve CLI tests (reference bin/cbresolve behaviors)."""

import asyncio

import pytest

from cueball_amd.cli import main, parse_ip_port, parse_time_interval


def test_parse_ip_port():
    assert parse_ip_port("1.2.3.4:80") == {"address": "1.2.3.4", "port": 80}
    with pytest.raises(ValueError):
        parse_ip_port("1.2.3.4")
    with pytest.raises(ValueError):
        parse_ip_port("nothost:80")


def test_parse_time_interval():
    assert parse_time_interval("5000") == 5000
    assert parse_time_interval("150ms") == 150
    assert parse_time_interval("5s") == 5000
    assert parse_time_interval("2m") == 120000
    with pytest.raises(ValueError):
        parse_time_interval("0")
    with pytest.raises(ValueError):
        parse_time_interval("5h")


def test_static_resolution(capsys):
    rc = main(["-S", "10.0.0.1:80", "10.0.0.2:8080"])
    assert rc == 0
    out = capsys.readouterr().out
    lines = [l for l in out.splitlines() if l.strip()]
    assert len(lines) == 2
    assert "10.0.0.1" in lines[0]
    assert "8080" in lines[1]


def test_static_bad_input(capsys):
    rc = main(["-S", "not-an-ip"])
    assert rc == 2


def test_dns_resolution_against_mock(capsys):
    """Full DNS path through the CLI against a local mock DNS server."""
    # spin up the mock server on a private loop first to learn its port
    from cueball_amd.testing import MockDnsServer

    async def serve_and_resolve():
        srv = MockDnsServer()
        await srv.start()
        srv.add_srv("_http._tcp.svc.test", "b1.svc.test", 8080)
        srv.add_a("b1.svc.test", "127.0.0.99")

        from cueball_amd.resolver import DNSResolverFSM
        DNSResolverFSM._nic_cache = {"lo": [
            {"family": "IPv4", "address": "127.0.0.1"}]}
        DNSResolverFSM._nic_cache_updated = \
            asyncio.get_running_loop().time() * 1000.0
        return srv

    # The CLI runs its own asyncio.run(); emulate by running the whole
    # flow in one loop via its underlying pieces is already covered in
    # test_dns_client; here just verify the argument plumbing end-to-end
    # with a failing (unreachable) resolver and a short timeout.
    rc = main(["-r", "127.0.0.1@1", "-t", "200ms", "-p", "80",
               "doesnot.exist.test"])
    assert rc == 1  # resolver failed -> exit 1 (reference cbrFailed)


def test_cli_dns_end_to_end_subprocess():
    """bin/cbresolve resolves through a real mock DNS server."""
    import asyncio
    import os
    import subprocess
    import sys

    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    async def serve():
        from cueball_amd.testing import MockDnsServer
        srv = MockDnsServer()
        await srv.start()
        srv.add_srv("_web._tcp.shop.test", "s1.shop.test", 8443, ttl=60)
        srv.add_a("s1.shop.test", "10.9.8.7", ttl=60)
        proc = await asyncio.create_subprocess_exec(
            sys.executable, os.path.join(ROOT, "bin", "cbresolve"),
            "-r", srv.resolver_address, "-s", "_web._tcp",
            "-t", "5s", "shop.test",
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE, cwd=ROOT)
        out, err = await asyncio.wait_for(proc.communicate(), timeout=60)
        srv.stop()
        return proc.returncode, out.decode(), err.decode()

    loop = asyncio.new_event_loop()
    try:
        rc, out, err = loop.run_until_complete(serve())
    finally:
        loop.close()
    assert rc == 0, (out, err)
    assert "10.9.8.7" in out
    assert "8443" in out


def test_cli_kang_listener():
    """cbresolve -k serves /kang/snapshot while following."""
    import asyncio
    import json as mod_json
    import os
    import socket
    import subprocess
    import sys
    import urllib.request

    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    kang_port = s.getsockname()[1]
    s.close()

    async def serve():
        from cueball_amd.testing import MockDnsServer
        srv = MockDnsServer()
        await srv.start()
        srv.add_srv("_web._tcp.shop.test", "s1.shop.test", 8443, ttl=60)
        srv.add_a("s1.shop.test", "10.9.8.7", ttl=60)
        proc = await asyncio.create_subprocess_exec(
            sys.executable, os.path.join(ROOT, "bin", "cbresolve"),
            "-r", srv.resolver_address, "-s", "_web._tcp",
            "-f", "-k", str(kang_port), "shop.test",
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE, cwd=ROOT)
        try:
            # wait for the follow-mode process to resolve + serve kang
            data = None
            for _ in range(100):
                await asyncio.sleep(0.1)
                try:
                    with urllib.request.urlopen(
                            "http://127.0.0.1:%d/kang/snapshot"
                            % kang_port, timeout=2) as r:
                        data = mod_json.loads(r.read())
                    if data.get("dns_res"):
                        break
                except OSError:
                    continue
            assert data is not None
            assert data["service"]["name"] == "cueball"
            assert len(data["dns_res"]) >= 1
        finally:
            proc.terminate()
            await proc.wait()
            srv.stop()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(serve())
    finally:
        loop.close()

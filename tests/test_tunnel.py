"""Tunnel plumbing against a stub cloudflared binary."""

import stat
import textwrap

import pytest

from rllm_amd.gateway.tunnel import (CloudflaredTunnel, TunnelError,
                                     container_reachable_url, public_gateway_url)


def _stub_cloudflared(tmp_path, body: str) -> str:
    p = tmp_path / "cloudflared"
    p.write_text("#!/bin/bash\n" + textwrap.dedent(body))
    p.chmod(p.stat().st_mode | stat.S_IEXEC)
    return str(p)


def test_container_reachable_url():
    assert container_reachable_url("http://127.0.0.1:8089/sessions/x/v1") == \
        "http://host.docker.internal:8089/sessions/x/v1"
    assert container_reachable_url("http://localhost:1/v1") == "http://host.docker.internal:1/v1"
    assert container_reachable_url("http://10.0.0.5:1/v1") == "http://10.0.0.5:1/v1"


def test_tunnel_parses_url_and_closes(tmp_path):
    binary = _stub_cloudflared(tmp_path, """
        echo "INF Starting tunnel"
        echo "INF +  https://brave-fox-abc123.trycloudflare.com  +"
        sleep 60
    """)
    t = CloudflaredTunnel("http://127.0.0.1:9", binary=binary, startup_timeout=10.0)
    url = t.start()
    assert url == "https://brave-fox-abc123.trycloudflare.com"
    t.close()
    assert t._proc is None


def test_tunnel_timeout_when_no_url(tmp_path):
    binary = _stub_cloudflared(tmp_path, 'echo "no url here"\n')
    t = CloudflaredTunnel("http://127.0.0.1:9", binary=binary, startup_timeout=2.0)
    with pytest.raises(TunnelError):
        t.start()


def test_tunnel_missing_binary():
    t = CloudflaredTunnel("http://127.0.0.1:9", binary="/does/not/exist")
    with pytest.raises(TunnelError):
        t.start()


def test_public_gateway_url_local_backend():
    url, tunnel = public_gateway_url("http://127.0.0.1:8089", "docker")
    assert tunnel is None
    assert url.startswith("http://host.docker.internal")


def test_public_gateway_url_remote_backend(tmp_path):
    binary = _stub_cloudflared(tmp_path, """
        echo "https://x-y-z.trycloudflare.com"
        sleep 60
    """)
    url, tunnel = public_gateway_url("http://127.0.0.1:8089", "modal", binary=binary)
    assert url == "https://x-y-z.trycloudflare.com"
    tunnel.close()

"""/admin/reload authorization (round-1 advisor medium finding: the model
hot-swap endpoint deserializes an arbitrary on-disk path and must not be
reachable by arbitrary clients of the public listener)."""

from __future__ import annotations

from creditcore.config import ServeConfig
from creditcore.serve import admin_authorized


def _cfg(token=""):
    c = ServeConfig()
    c.admin_token = token
    return c


def test_no_token_loopback_only():
    cfg = _cfg()
    assert admin_authorized(cfg, "127.0.0.1")
    assert admin_authorized(cfg, "::1")
    assert admin_authorized(cfg, "testclient")  # in-process ASGI test client
    assert admin_authorized(cfg, None)
    assert not admin_authorized(cfg, "10.0.0.7")
    assert not admin_authorized(cfg, "203.0.113.9")


def test_token_required_everywhere_when_set():
    cfg = _cfg("s3cret")
    # loopback no longer suffices once a token is configured
    assert not admin_authorized(cfg, "127.0.0.1")
    assert admin_authorized(cfg, "10.0.0.7", x_admin_token="s3cret")
    assert admin_authorized(cfg, "10.0.0.7", authorization="Bearer s3cret")
    assert not admin_authorized(cfg, "10.0.0.7", authorization="Bearer nope")
    assert not admin_authorized(cfg, "10.0.0.7", x_admin_token="wrong")


def test_fastapi_reload_gated(model_dir):
    from fastapi.testclient import TestClient

    from creditcore.serve import create_app

    cfg = _cfg("tok")
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    with TestClient(create_app(cfg)) as client:
        r = client.post("/admin/reload", json={"model_uri": model_dir})
        assert r.status_code == 403
        r = client.post(
            "/admin/reload",
            json={"model_uri": model_dir},
            headers={"X-Admin-Token": "tok"},
        )
        assert r.status_code == 200
        # scoring unaffected by the auth layer
        from creditcore.schema import SAMPLE_REQUEST

        assert client.post("/score", json=SAMPLE_REQUEST).status_code == 200


def test_fastapi_reload_rejects_remote_client_without_token(model_dir):
    """No token configured: a non-loopback client address gets 403 on
    /admin/reload while /score keeps working (the actual reported hole —
    0.0.0.0 public listener, any remote client could hot-swap the model)."""
    import anyio
    import httpx

    from creditcore.serve import create_app

    cfg = _cfg()  # no token -> loopback-only
    cfg.model_directory = model_dir
    cfg.device = "cpu"
    app = create_app(cfg)

    async def go():
        async with app.router.lifespan_context(app):
            transport = httpx.ASGITransport(app, client=("203.0.113.9", 4242))
            async with httpx.AsyncClient(
                transport=transport, base_url="http://svc"
            ) as c:
                r = await c.post("/admin/reload", json={"model_uri": model_dir})
                assert r.status_code == 403
                from creditcore.schema import SAMPLE_REQUEST

                ok = await c.post("/score", json=SAMPLE_REQUEST)
                assert ok.status_code == 200

    anyio.run(go)



def test_workload_identity_token_exchange(tmp_path, monkeypatch):
    """AKS workload-identity: the projected federated token is exchanged
    at the (mocked) AAD endpoint via the client_assertion grant, and blob
    downloads carry the bearer header. SAS, when present, wins."""
    import http.server
    import json
    import threading

    from kaito_amd.models import streaming

    seen = {}

    class Handler(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            body = self.rfile.read(int(self.headers["Content-Length"]))
            seen["body"] = body.decode()
            seen["path"] = self.path
            out = json.dumps({"access_token": "tok-123",
                              "expires_in": 3600}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(out)))
            self.end_headers()
            self.wfile.write(out)

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        tf = tmp_path / "federated.jwt"
        tf.write_text("fed-jwt-abc")
        monkeypatch.setenv("AZURE_FEDERATED_TOKEN_FILE", str(tf))
        monkeypatch.setenv("AZURE_CLIENT_ID", "cid")
        monkeypatch.setenv("AZURE_TENANT_ID", "tid")
        monkeypatch.setenv("AZURE_AUTHORITY_HOST",
                           f"http://127.0.0.1:{srv.server_address[1]}")
        monkeypatch.delenv("AZURE_STORAGE_SAS_TOKEN", raising=False)

        tok = streaming.azure_workload_identity_token()
        assert tok == "tok-123"
        assert seen["path"] == "/tid/oauth2/v2.0/token"
        assert "client_assertion=fed-jwt-abc" in seen["body"]
        assert "client_id=cid" in seen["body"]
        assert "jwt-bearer" in seen["body"]

        hdr = streaming.azure_auth_headers()
        assert hdr["Authorization"] == "Bearer tok-123"

        # SAS present → URL auth, no bearer
        monkeypatch.setenv("AZURE_STORAGE_SAS_TOKEN", "sv=x&sig=y")
        assert streaming.azure_auth_headers() == {}
    finally:
        srv.shutdown()


def test_workload_identity_absent_returns_none(monkeypatch):
    from kaito_amd.models import streaming
    for k in ("AZURE_FEDERATED_TOKEN_FILE", "AZURE_CLIENT_ID",
              "AZURE_TENANT_ID", "AZURE_STORAGE_SAS_TOKEN"):
        monkeypatch.delenv(k, raising=False)
    assert streaming.azure_workload_identity_token() is None
    assert streaming.azure_auth_headers() == {}


def test_concurrent_fetch_sharded_checkpoint(tmp_path, monkeypatch):
    """Sharded safetensors fetch runs KAITO_STREAM_CONCURRENCY parallel
    connections and aggregates progress monotonically to 1.0."""
    import functools
    import http.server
    import threading

    from kaito_amd.models.streaming import fetch_weights

    src = tmp_path / "src"
    src.mkdir()
    names = [f"model-{i:05d}-of-00004.safetensors" for i in range(1, 5)]
    for nm in names:
        (src / nm).write_bytes(bytes([len(nm) % 251]) * 65536)

    handler = functools.partial(
        http.server.SimpleHTTPRequestHandler, directory=str(src))
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        monkeypatch.setenv("KAITO_STREAM_CONCURRENCY", "4")
        seen = []
        dest = tmp_path / "cache"
        got = fetch_weights(
            f"http://127.0.0.1:{srv.server_address[1]}",
            cache_dir=str(dest), files=names, progress=seen.append)
        for nm in names:
            assert (dest / nm).read_bytes() == (src / nm).read_bytes()
        assert got == str(dest)
        assert seen and abs(seen[-1] - 1.0) < 1e-6
    finally:
        srv.shutdown()

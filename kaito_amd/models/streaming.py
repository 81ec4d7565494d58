"""Model weight streaming — the MI355X analog of the reference's
model-streaming subsystem (pkg/workspace/inference/modelstreaming: az://
path resolution, SAS-token auth via fetch_sas.py init container, RunAI
streamer --load-format=runai_streamer; SURVEY.md §2 "Model streaming" row).

Sources:
  file://path or plain path  → used in place
  http(s)://...              → chunked streaming download with progress
                               callbacks (drives kaito_model_download_*)
  az://account/container/blob → resolved to the blob HTTPS endpoint;
                               auth is SAS (AZURE_STORAGE_SAS_TOKEN, the
                               fetch-sas init-container contract) or AKS
                               WORKLOAD IDENTITY: the projected federated
                               token (AZURE_FEDERATED_TOKEN_FILE +
                               AZURE_CLIENT_ID + AZURE_TENANT_ID) is
                               exchanged at AAD for a storage bearer
                               token (client_assertion grant) and sent as
                               Authorization: Bearer
Downloads go to a local cache dir (the NVMe PVC mount in-cluster).
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Callable, List, Optional
from urllib.parse import urlparse

ProgressCb = Callable[[float], None]   # 0.0 - 1.0


def resolve_azure_url(url: str) -> str:
    """az://account/container/path → https://account.blob.core.windows.net/
    container/path[?sas] (reference: sasblob.go path resolution)."""
    p = urlparse(url)
    account = p.netloc
    path = p.path.lstrip("/")
    base = f"https://{account}.blob.core.windows.net/{path}"
    sas = os.environ.get("AZURE_STORAGE_SAS_TOKEN", "")
    if sas:
        base += ("" if sas.startswith("?") else "?") + sas
    return base


def azure_workload_identity_token(
        scope: str = "https://storage.azure.com/.default") -> Optional[str]:
    """AKS workload-identity federation: exchange the projected service-
    account token for an AAD access token (the reference streams az://
    weights under workload identity when no SAS is provisioned). Returns
    None when the identity env is absent. AZURE_AUTHORITY_HOST overrides
    the login endpoint (tests point it at a local mock)."""
    token_file = os.environ.get("AZURE_FEDERATED_TOKEN_FILE")
    client_id = os.environ.get("AZURE_CLIENT_ID")
    tenant = os.environ.get("AZURE_TENANT_ID")
    if not (token_file and client_id and tenant):
        return None
    import httpx
    assertion = Path(token_file).read_text().strip()
    authority = os.environ.get("AZURE_AUTHORITY_HOST",
                               "https://login.microsoftonline.com")
    r = httpx.post(
        f"{authority.rstrip('/')}/{tenant}/oauth2/v2.0/token",
        data={
            "grant_type": "client_credentials",
            "client_id": client_id,
            "scope": scope,
            "client_assertion_type":
                "urn:ietf:params:oauth:client-assertion-type:jwt-bearer",
            "client_assertion": assertion,
        }, timeout=30)
    r.raise_for_status()
    return r.json()["access_token"]


def azure_auth_headers() -> dict:
    """Bearer headers for blob GETs when workload identity is active and
    no SAS token is configured (SAS rides the URL instead)."""
    if os.environ.get("AZURE_STORAGE_SAS_TOKEN"):
        return {}
    tok = azure_workload_identity_token()
    if tok is None:
        return {}
    return {"Authorization": f"Bearer {tok}", "x-ms-version": "2021-08-06"}


def _download_http(url: str, dest: Path, progress: Optional[ProgressCb],
                   chunk_bytes: int = 8 << 20,
                   headers: Optional[dict] = None) -> Path:
    import httpx
    dest.parent.mkdir(parents=True, exist_ok=True)
    tmp = dest.with_suffix(dest.suffix + ".part")
    with httpx.stream("GET", url, follow_redirects=True, timeout=600,
                      headers=headers or {}) as r:
        r.raise_for_status()
        total = int(r.headers.get("content-length", 0)) or None
        got = 0
        with open(tmp, "wb") as f:
            for chunk in r.iter_bytes(chunk_bytes):
                f.write(chunk)
                got += len(chunk)
                if progress and total:
                    progress(min(got / total, 1.0))
    tmp.rename(dest)
    if progress:
        progress(1.0)
    return dest


def fetch_weights(source: str, cache_dir: str = "/workspace/weights",
                  files: Optional[List[str]] = None,
                  progress: Optional[ProgressCb] = None) -> str:
    """Materialize a weights directory from `source`; returns a local path
    suitable for load_safetensors_weights()."""
    p = urlparse(source)
    if p.scheme in ("", "file"):
        path = p.path if p.scheme == "file" else source
        if not os.path.exists(path):
            raise FileNotFoundError(path)
        if progress:
            progress(1.0)
        return path
    headers = None
    if p.scheme == "az":
        source = resolve_azure_url(source)
        p = urlparse(source)
        headers = azure_auth_headers() or None
    if p.scheme in ("http", "https"):
        base = source.rstrip("/")
        names = files or ["model.safetensors", "config.json",
                          "tokenizer.json", "tokenizer_config.json"]
        out = Path(cache_dir)
        n = len(names)
        # concurrent streaming (the RunAI-streamer analog: KAITO_STREAM_
        # CONCURRENCY parallel connections; sharded checkpoints with many
        # model-xxxxx-of-yyyyy.safetensors files saturate NVMe this way)
        workers = max(1, int(os.environ.get("KAITO_STREAM_CONCURRENCY",
                                            "4")))
        fracs = [0.0] * n
        lock = __import__("threading").Lock()

        def fetch_one(i: int, name: str):
            def sub(frac, i=i):
                if progress:
                    with lock:
                        fracs[i] = frac
                        progress(sum(fracs) / n)
            try:
                _download_http(f"{base}/{name}", out / name, sub,
                               headers=headers)
            except Exception:  # noqa: BLE001 — optional aux files
                if name.endswith(".safetensors"):
                    raise

        if workers == 1 or n == 1:
            for i, name in enumerate(names):
                fetch_one(i, name)
        else:
            from concurrent.futures import ThreadPoolExecutor
            with ThreadPoolExecutor(max_workers=workers) as ex:
                list(ex.map(lambda t: fetch_one(*t), enumerate(names)))
        return str(out)
    raise ValueError(f"unsupported weight source scheme {p.scheme!r}")

"""RAGEngine tests: vector store, BM25, hybrid fusion, persistence, and
the FastAPI endpoint surface (reference: presets/ragengine/tests)."""
import numpy as np
import pytest
from fastapi.testclient import TestClient

from kaito_amd.ragengine.bm25 import BM25Index
from kaito_amd.ragengine.config import RagConfig
from kaito_amd.ragengine.embeddings import HashEmbedding
from kaito_amd.ragengine.service import build_rag_app
from kaito_amd.ragengine.vector_store import (FlatIndex, VectorStoreIndex,
                                              VectorStoreManager, doc_id_for)

DOCS = [
    "the quick brown fox jumps over the lazy dog",
    "kubernetes operators reconcile desired state",
    "mi355x has 288 gigabytes of hbm3e memory",
    "paged attention stores kv cache in fixed blocks",
    "a fox is a small wild canine animal",
]


@pytest.fixture
def index():
    idx = VectorStoreIndex("t", HashEmbedding(), use_gpu=False)
    idx.index_documents(DOCS)
    return idx


def test_doc_id_is_text_hash():
    assert doc_id_for("abc") == doc_id_for("abc")
    assert doc_id_for("abc") != doc_id_for("abd")
    assert len(doc_id_for("abc")) == 64


def test_flat_index_add_remove_search():
    f = FlatIndex(4, use_gpu=False)
    f.add("a", np.array([1, 0, 0, 0], dtype=np.float32))
    f.add("b", np.array([0, 1, 0, 0], dtype=np.float32))
    f.add("c", np.array([0.9, 0.1, 0, 0], dtype=np.float32))
    hits = f.search(np.array([1, 0, 0, 0], dtype=np.float32), 2)
    assert hits[0][0] == "a" and hits[1][0] == "c"
    f.remove("a")
    hits = f.search(np.array([1, 0, 0, 0], dtype=np.float32), 2)
    assert hits[0][0] == "c"
    assert len(f) == 2


def test_bm25_ranks_term_matches():
    b = BM25Index()
    for i, d in enumerate(DOCS):
        b.add(str(i), d)
    hits = b.search("fox", top_k=3)
    ids = [h[0] for h in hits]
    assert set(ids[:2]) == {"0", "4"}
    b.remove("0")
    hits = b.search("fox", 3)
    assert hits[0][0] == "4"


def test_hybrid_retrieval_prefers_exact_terms(index):
    out = index.retrieve("fox jumps", top_k=3)
    assert out
    assert "fox" in out[0].text


def test_update_and_delete(index):
    did = doc_id_for(DOCS[0])
    new_id = index.update_document(did, "completely new text about cats")
    assert new_id != did
    assert index.delete_document(new_id)
    assert not index.delete_document("nonexistent")


def test_persist_load_roundtrip(tmp_path, index):
    index.persist(str(tmp_path / "t"))
    idx2 = VectorStoreIndex("t", HashEmbedding(), use_gpu=False)
    idx2.load(str(tmp_path / "t"))
    assert len(idx2.docs) == len(DOCS)
    out = idx2.retrieve("kubernetes operators", top_k=1)
    assert "kubernetes" in out[0].text


# ------------------------------------------------------------ service
@pytest.fixture
def client(tmp_path):
    cfg = RagConfig()
    cfg.persist_dir = str(tmp_path)
    emb = HashEmbedding()
    app = build_rag_app(cfg, emb, VectorStoreManager(emb, use_gpu=False))
    return TestClient(app)


def test_service_index_and_retrieve(client):
    r = client.post("/index", json={
        "index_name": "kb",
        "documents": [{"text": d} for d in DOCS]})
    assert r.status_code == 200
    assert len(r.json()) == len(DOCS)
    assert client.get("/indexes").json() == ["kb"]
    r = client.post("/retrieve", json={"index_name": "kb",
                                       "query": "kv cache blocks", "top_k": 2})
    assert r.status_code == 200
    res = r.json()["results"]
    assert res and "kv cache" in res[0]["text"]


def test_service_documents_crud(client):
    client.post("/index", json={"index_name": "kb",
                                "documents": [{"text": "hello world"}]})
    docs = client.get("/indexes/kb/documents").json()
    assert docs["count"] == 1
    did = docs["documents"][0]["doc_id"]
    r = client.post(f"/indexes/kb/documents/{did}",
                    json={"text": "goodbye world"})
    assert r.status_code == 200
    r = client.delete(f"/indexes/kb/documents/{r.json()['doc_id']}")
    assert r.status_code == 200
    assert client.get("/indexes/kb/documents").json()["count"] == 0


def test_service_persist_load(client):
    client.post("/index", json={"index_name": "kb",
                                "documents": [{"text": "persist me"}]})
    assert client.post("/persist/kb").status_code == 200
    assert client.delete("/indexes/kb").status_code == 200
    assert client.post("/load/kb").status_code == 200
    r = client.post("/retrieve", json={"index_name": "kb",
                                       "query": "persist", "top_k": 1})
    assert r.json()["results"][0]["text"] == "persist me"


def test_service_404s(client):
    assert client.post("/retrieve", json={
        "index_name": "nope", "query": "x"}).status_code == 404
    assert client.get("/indexes/nope/documents").status_code == 404
    assert client.delete("/indexes/nope").status_code == 404


def test_service_health_metrics(client):
    assert client.get("/health").json() == {"status": "ok"}
    client.post("/index", json={"index_name": "kb",
                                "documents": [{"text": "m"}]})
    assert "kaito_rag_request_latency_seconds" in client.get("/metrics").text


# ------------------------------------------------------------ guardrails
def test_guardrails_scan_block_and_redact(tmp_path):
    from kaito_amd.ragengine.guardrails import PolicyLoader, Scanner
    pol = tmp_path / "policy.yaml"
    pol.write_text("""
blocked_keywords: ["forbidden"]
redactions:
  - pattern: "\\\\b\\\\d{3}-\\\\d{2}-\\\\d{4}\\\\b"
    replacement: "[SSN]"
""")
    sc = Scanner(PolicyLoader(str(pol)))
    res = sc.scan("my ssn is 123-45-6789 ok")
    assert res.ok and "[SSN]" in res.text and "123-45" not in res.text
    res = sc.scan("this mentions Forbidden things")
    assert not res.ok and "blocked" in res.text


def test_guardrails_hot_reload(tmp_path):
    import os, time
    from kaito_amd.ragengine.guardrails import PolicyLoader, Scanner
    pol = tmp_path / "p.yaml"
    pol.write_text("blocked_keywords: []\n")
    sc = Scanner(PolicyLoader(str(pol), hot_reload=True))
    assert sc.scan("hello bad").ok
    pol.write_text("blocked_keywords: ['bad']\n")
    os.utime(pol, (time.time() + 5, time.time() + 5))
    assert not sc.scan("hello bad").ok


def test_buffer_window_catches_split_matches(tmp_path):
    from kaito_amd.ragengine.guardrails import (BufferWindowScanner,
                                                PolicyLoader, Scanner)
    pol = tmp_path / "p.yaml"
    pol.write_text("blocked_keywords: ['topsecret']\n")
    bw = BufferWindowScanner(Scanner(PolicyLoader(str(pol))), window=16)
    out = bw.feed("this is top")
    out += bw.feed("secret info and much more text to push the window")
    assert bw.blocked
    assert "blocked" in out
    # clean stream passes through
    bw2 = BufferWindowScanner(Scanner(PolicyLoader(str(pol))), window=8)
    text = "a perfectly clean sentence streaming through"
    got = ""
    for i in range(0, len(text), 7):
        got += bw2.feed(text[i:i + 7])
    got += bw2.flush()
    assert got == text


# ------------------------------------------------------------ qdrant backend
def _fake_qdrant_app():
    """Minimal in-process Qdrant REST double (collections/points API)."""
    from fastapi import FastAPI
    app = FastAPI()
    store = {}

    @app.get("/collections/{name}")
    def getc(name: str):
        from fastapi.responses import JSONResponse
        if name in store:
            return {"result": {}}
        return JSONResponse({"status": "not found"}, status_code=404)

    @app.put("/collections/{name}")
    def putc(name: str, body: dict):
        store[name] = {}
        return {"result": True}

    @app.put("/collections/{name}/points")
    def upsert(name: str, body: dict):
        for pt in body["points"]:
            store[name][pt["id"]] = pt
        return {"result": {}}

    @app.post("/collections/{name}/points/delete")
    def delete(name: str, body: dict):
        for pid in body["points"]:
            store[name].pop(pid, None)
        return {"result": {}}

    @app.post("/collections/{name}/points/search")
    def search(name: str, body: dict):
        q = np.array(body["vector"], dtype=np.float32)
        hits = []
        for pt in store[name].values():
            v = np.array(pt["vector"], dtype=np.float32)
            denom = (np.linalg.norm(q) * np.linalg.norm(v)) or 1.0
            hits.append({"id": pt["id"], "score": float(q @ v / denom),
                         "payload": pt["payload"]})
        hits.sort(key=lambda h: -h["score"])
        return {"result": hits[: body["limit"]]}

    return app


def test_qdrant_index_backend():
    import socket
    import threading
    import time
    import uvicorn
    from kaito_amd.ragengine.qdrant_store import QdrantIndex
    app = _fake_qdrant_app()
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                           log_level="error"))
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    deadline = time.monotonic() + 10
    while not server.started and time.monotonic() < deadline:
        time.sleep(0.05)
    idx = QdrantIndex(4, "kb", f"http://127.0.0.1:{port}")
    idx.add("a", np.array([1, 0, 0, 0], dtype=np.float32))
    idx.add("b", np.array([0, 1, 0, 0], dtype=np.float32))
    idx.add("c", np.array([0.9, 0.1, 0, 0], dtype=np.float32))
    hits = idx.search(np.array([1, 0, 0, 0], dtype=np.float32), 2)
    assert [h[0] for h in hits] == ["a", "c"]
    idx.remove("a")
    hits = idx.search(np.array([1, 0, 0, 0], dtype=np.float32), 2)
    assert hits[0][0] == "c"
    assert len(idx) == 2
    server.should_exit = True


def test_ivf_index_recall_and_persistence(tmp_path):
    """IVF-flat: before training it matches flat scan exactly; after
    training, nprobe=nlist is exact and a moderate nprobe keeps high
    recall@10 on clustered data. Persistence reuses the flat state path."""
    import numpy as np
    from kaito_amd.ragengine.ivf import IVFFlatIndex
    from kaito_amd.ragengine.vector_store import FlatIndex

    rng = np.random.default_rng(3)
    dim, n_clusters, per = 32, 16, 40
    centers = rng.normal(size=(n_clusters, dim))
    vecs, ids = [], []
    for c in range(n_clusters):
        pts = centers[c] + 0.05 * rng.normal(size=(per, dim))
        for j, p in enumerate(pts):
            p = p / np.linalg.norm(p)
            vecs.append(p.astype(np.float32))
            ids.append(f"c{c}_{j}")

    flat = FlatIndex(dim, use_gpu=False)
    exact = IVFFlatIndex(dim, nlist=16, nprobe=16, min_train=10**9,
                         use_gpu=False)
    ivf = IVFFlatIndex(dim, nlist=16, nprobe=4, min_train=len(ids),
                       use_gpu=False)
    for d, v in zip(ids, vecs):
        flat.add(d, v)
        exact.add(d, v)
        ivf.add(d, v)

    q = vecs[7] + 0.01 * rng.normal(size=dim)
    q = (q / np.linalg.norm(q)).astype(np.float32)
    # untrained IVF == flat scan, exactly
    assert [d for d, _ in exact.search(q, 10)] == \
        [d for d, _ in flat.search(q, 10)]
    assert ivf.centroids is not None       # auto-trained at min_train
    truth = {d for d, _ in flat.search(q, 10)}
    got = {d for d, _ in ivf.search(q, 10)}
    assert len(truth & got) >= 8           # recall@10 ≥ 0.8 at nprobe=4

    # full-probe IVF is exact (same set; scores equal)
    ivf.nprobe = 16
    full = {d for d, _ in ivf.search(q, 10)}
    assert full == truth

    # persistence via the VectorStoreIndex path shape (ids + vecs)
    st = ivf.state()
    re = IVFFlatIndex(dim, nlist=16, nprobe=16, min_train=len(ids),
                      use_gpu=False)
    re.load_state(st["ids"], st["vecs"], st["centroids"])
    assert {d for d, _ in re.search(q, 10)} == truth

    # removal keeps the structures consistent
    victim = next(iter(truth))
    ivf.remove(victim)
    assert victim not in {d for d, _ in ivf.search(q, 10)}


def test_ivf_backend_selected_by_env(monkeypatch):
    monkeypatch.setenv("VECTOR_DB_TYPE", "ivf")
    monkeypatch.setenv("IVF_MIN_TRAIN", "1000000")
    from kaito_amd.ragengine.config import RagConfig
    from kaito_amd.ragengine.service import build_rag_app
    from kaito_amd.ragengine.ivf import IVFFlatIndex
    from fastapi.testclient import TestClient
    app = build_rag_app(RagConfig())
    with TestClient(app) as c:
        r = c.post("/index", json={"index_name": "kb",
                                   "documents": [{"text": "alpha beta"},
                                                 {"text": "gamma delta"}]})
        assert r.status_code == 200
        idx = app.state.manager.get("kb")
        assert isinstance(idx.flat, IVFFlatIndex)
        r = c.post("/retrieve", json={"index_name": "kb",
                                      "query": "alpha", "top_k": 1})
        assert r.status_code == 200

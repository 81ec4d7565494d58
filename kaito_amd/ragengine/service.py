"""RAGEngine FastAPI service.

Endpoint surface kept parity with the reference's presets/ragengine/main.py
(grep @app. main.py:106-876): /index, /indexes, /indexes/{name}/documents
(list/update/delete), /persist/{name}, /load/{name}, /retrieve,
/v1/chat/completions (context-injecting passthrough to the Workspace LLM,
incl. SSE streaming), /metrics, /health.
"""
from __future__ import annotations

import json
import time
from typing import Any, Dict, List, Optional

import httpx
from fastapi import FastAPI, HTTPException
from fastapi.responses import Response, StreamingResponse
from prometheus_client import (CollectorRegistry, Counter, Histogram,
                               generate_latest, CONTENT_TYPE_LATEST)
from pydantic import BaseModel, Field

from .config import RagConfig
from .embeddings import make_embedding
from .guardrails import BufferWindowScanner, PolicyLoader, Scanner
from .vector_store import VectorStoreManager

RAG_REGISTRY = CollectorRegistry()
REQ_LATENCY = Histogram("kaito_rag_request_latency_seconds",
                        "Per-endpoint latency", ["endpoint"],
                        registry=RAG_REGISTRY)
REQ_COUNT = Counter("kaito_rag_requests", "Requests", ["endpoint", "status"],
                    registry=RAG_REGISTRY)


class IndexRequest(BaseModel):
    index_name: str
    documents: List[Dict[str, Any]]  # {text, metadata?}


class RetrieveRequest(BaseModel):
    index_name: str
    query: str
    top_k: int = 5


class ChatRequest(BaseModel):
    model: str = ""
    messages: List[Dict[str, str]] = Field(default_factory=list)
    index_name: Optional[str] = None
    context_token_ratio: float = 0.5
    max_tokens: Optional[int] = None
    temperature: float = 0.7
    stream: bool = False


def _sse_text_chunk(text: str) -> str:
    payload = {"object": "chat.completion.chunk",
               "choices": [{"index": 0, "delta": {"content": text},
                            "finish_reason": None}]}
    return f"data: {json.dumps(payload)}\n\n"


def build_rag_app(cfg: Optional[RagConfig] = None, embedding=None,
                  manager: Optional[VectorStoreManager] = None) -> FastAPI:
    cfg = cfg or RagConfig()
    embedding = embedding or make_embedding(cfg)
    if manager is None:
        factory = None
        kind = cfg.vector_db_type.lower()
        if kind == "qdrant" and cfg.vector_db_url:
            from .qdrant_store import make_index
            factory = lambda dim, name: make_index(cfg, dim, name)  # noqa: E731
        elif kind == "ivf":
            from .ivf import IVFFlatIndex
            factory = lambda dim, name: IVFFlatIndex(  # noqa: E731
                dim, cfg.ivf_nlist, cfg.ivf_nprobe, cfg.ivf_min_train)
        manager = VectorStoreManager(embedding, index_factory=factory)
    app = FastAPI(title="kaito-amd ragengine")
    app.state.manager = manager
    app.state.cfg = cfg
    scanner = None
    if cfg.guardrails_enabled:
        scanner = Scanner(PolicyLoader(cfg.guardrails_policy_path,
                                       cfg.guardrails_hot_reload))
    app.state.scanner = scanner

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def metrics():
        return Response(generate_latest(RAG_REGISTRY),
                        media_type=CONTENT_TYPE_LATEST)

    # ------------------------------------------------------------- indexing
    @app.post("/index")
    async def index(req: IndexRequest):
        t0 = time.monotonic()
        idx = manager.get(req.index_name, create=True)
        texts = [d["text"] for d in req.documents]
        metas = [d.get("metadata", {}) for d in req.documents]
        ids = idx.index_documents(texts, metas)
        REQ_LATENCY.labels("index").observe(time.monotonic() - t0)
        REQ_COUNT.labels("index", "200").inc()
        return [{"doc_id": i, "text": t} for i, t in zip(ids, texts)]

    @app.get("/indexes")
    async def indexes():
        return manager.list_indexes()

    @app.get("/indexes/{name}/documents")
    async def documents(name: str, limit: int = 100, offset: int = 0):
        try:
            idx = manager.get(name)
        except KeyError:
            raise HTTPException(404, f"index {name} not found")
        return {"documents": idx.list_documents(limit, offset),
                "count": len(idx.docs)}

    @app.post("/indexes/{name}/documents/{doc_id}")
    async def update_document(name: str, doc_id: str, body: Dict[str, Any]):
        try:
            idx = manager.get(name)
        except KeyError:
            raise HTTPException(404, f"index {name} not found")
        new_id = idx.update_document(doc_id, body["text"],
                                     body.get("metadata"))
        return {"doc_id": new_id}

    @app.delete("/indexes/{name}/documents/{doc_id}")
    async def delete_document(name: str, doc_id: str):
        try:
            idx = manager.get(name)
        except KeyError:
            raise HTTPException(404, f"index {name} not found")
        if not idx.delete_document(doc_id):
            raise HTTPException(404, f"document {doc_id} not found")
        return {"deleted": doc_id}

    @app.delete("/indexes/{name}")
    async def delete_index(name: str):
        if not manager.delete_index(name):
            raise HTTPException(404, f"index {name} not found")
        return {"deleted": name}

    @app.post("/persist/{name}")
    async def persist(name: str, path: Optional[str] = None):
        try:
            idx = manager.get(name)
        except KeyError:
            raise HTTPException(404, f"index {name} not found")
        import os
        dest = path or cfg.persist_dir
        idx.persist(os.path.join(dest, name))
        return {"persisted": name, "path": dest}

    @app.post("/load/{name}")
    async def load(name: str, path: Optional[str] = None):
        import os
        src = path or cfg.persist_dir
        try:
            manager.load_index(name, src)
        except FileNotFoundError:
            raise HTTPException(404, f"no persisted index at {src}/{name}")
        return {"loaded": name}

    # ------------------------------------------------------------ retrieval
    @app.post("/retrieve")
    async def retrieve(req: RetrieveRequest):
        t0 = time.monotonic()
        try:
            idx = manager.get(req.index_name)
        except KeyError:
            raise HTTPException(404, f"index {req.index_name} not found")
        docs = idx.retrieve(req.query, req.top_k,
                            cfg.vector_weight, cfg.bm25_weight)
        REQ_LATENCY.labels("retrieve").observe(time.monotonic() - t0)
        return {"results": [{"doc_id": d.doc_id, "text": d.text,
                             "score": d.score, "metadata": d.metadata}
                            for d in docs]}

    # ---------------------------------------------------- chat interception
    def _build_context(req: ChatRequest) -> Optional[str]:
        if not req.index_name:
            return None
        try:
            idx = manager.get(req.index_name)
        except KeyError:
            return None
        query = next((m["content"] for m in reversed(req.messages)
                      if m.get("role") == "user"), "")
        docs = idx.retrieve(query, 5, cfg.vector_weight, cfg.bm25_weight)
        # context-window budget (reference: LLM_CONTEXT_WINDOW, config.py:66)
        budget_chars = int(cfg.llm_context_window *
                           req.context_token_ratio) * 4
        parts, used = [], 0
        for d in docs:
            if used + len(d.text) > budget_chars:
                break
            parts.append(d.text)
            used += len(d.text)
        return "\n\n".join(parts) if parts else None

    @app.post("/v1/chat/completions")
    async def chat(req: ChatRequest):
        ctx = _build_context(req)
        messages = list(req.messages)
        if ctx:
            messages = [{"role": "system",
                         "content": "Use the following context to answer:\n"
                                    + ctx}] + messages
        payload = {"model": req.model, "messages": messages,
                   "temperature": req.temperature, "stream": req.stream}
        if req.max_tokens:
            payload["max_tokens"] = req.max_tokens
        headers = {}
        if cfg.llm_access_secret:
            headers["Authorization"] = f"Bearer {cfg.llm_access_secret}"
        if req.stream:
            async def relay():
                bw = BufferWindowScanner(scanner) if scanner else None
                async with httpx.AsyncClient(timeout=300) as client:
                    async with client.stream("POST", cfg.llm_inference_url,
                                             json=payload,
                                             headers=headers) as r:
                        async for line in r.aiter_lines():
                            if not line:
                                continue
                            if bw is None:
                                yield line + "\n\n"
                                continue
                            # guardrails: re-chunk SSE through the buffer
                            # window (reference streaming/guardrails.py)
                            if not line.startswith("data:") or \
                                    line.strip() == "data: [DONE]":
                                tail = bw.flush()
                                if tail:
                                    yield _sse_text_chunk(tail)
                                yield line + "\n\n"
                                continue
                            try:
                                obj = json.loads(line[5:].strip())
                                delta = obj["choices"][0].get(
                                    "delta", {}).get("content", "")
                            except (json.JSONDecodeError, KeyError,
                                    IndexError):
                                yield line + "\n\n"
                                continue
                            cleared = bw.feed(delta)
                            if bw.blocked:
                                yield _sse_text_chunk(cleared)
                                yield "data: [DONE]\n\n"
                                return
                            if cleared:
                                yield _sse_text_chunk(cleared)
            return StreamingResponse(relay(), media_type="text/event-stream")
        async with httpx.AsyncClient(timeout=300) as client:
            r = await client.post(cfg.llm_inference_url, json=payload,
                                  headers=headers)
            if scanner is not None and r.status_code == 200:
                try:
                    body = r.json()
                    msg = body["choices"][0]["message"]
                    res = scanner.scan(msg.get("content", ""))
                    msg["content"] = res.text
                    return Response(json.dumps(body),
                                    status_code=200,
                                    media_type="application/json")
                except (KeyError, IndexError, ValueError):
                    pass
            return Response(r.content, status_code=r.status_code,
                            media_type="application/json")

    return app


def main():
    import uvicorn
    app = build_rag_app()
    uvicorn.run(app, host="0.0.0.0", port=int(__import__("os").environ.get(
        "PORT", 5000)), log_level="warning")


if __name__ == "__main__":
    main()

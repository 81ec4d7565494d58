"""RAGEngine configuration — env contract kept byte-compatible with the
reference (presets/ragengine/config.py; wired from the CRD by
pkg/ragengine/manifests/manifests.go:155)."""
from __future__ import annotations

import os


def _env(name: str, default: str = "") -> str:
    return os.environ.get(name, default)


class RagConfig:
    def __init__(self):
        self.embedding_source_type = _env("EMBEDDING_SOURCE_TYPE", "local")
        self.local_embedding_model_id = _env("LOCAL_EMBEDDING_MODEL_ID",
                                             "BAAI/bge-small-en-v1.5")
        self.remote_embedding_url = _env("REMOTE_EMBEDDING_URL")
        self.remote_embedding_access_secret = _env("REMOTE_EMBEDDING_ACCESS_SECRET")
        self.vector_db_type = _env("VECTOR_DB_TYPE", "faiss")
        # faiss → local flat (GEMM + HIP top-k); ivf → local IVF-flat
        # (coarse-quantized, nprobe lists); qdrant → remote server
        self.ivf_nlist = int(_env("IVF_NLIST", "64"))
        self.ivf_nprobe = int(_env("IVF_NPROBE", "8"))
        self.ivf_min_train = int(_env("IVF_MIN_TRAIN", "256"))
        self.vector_db_url = _env("VECTOR_DB_URL")
        self.vector_db_access_secret = _env("VECTOR_DB_ACCESS_SECRET")
        self.llm_inference_url = _env("LLM_INFERENCE_URL",
                                      "http://localhost:5000/v1/chat/completions")
        self.llm_access_secret = _env("LLM_ACCESS_SECRET")
        self.llm_context_window = int(_env("LLM_CONTEXT_WINDOW", "8192"))
        self.persist_dir = _env("DEFAULT_VECTOR_DB_PERSIST_DIR",
                                "/tmp/kaito_rag_persist")
        self.guardrails_enabled = _env("OUTPUT_GUARDRAILS_ENABLED",
                                       "false").lower() == "true"
        self.guardrails_policy_path = _env("OUTPUT_GUARDRAILS_POLICY_PATH")
        self.guardrails_hot_reload = _env(
            "OUTPUT_GUARDRAILS_HOT_RELOAD_ENABLED", "false").lower() == "true"
        # hybrid fusion weights (reference: hybrid_retriever.py:132-166 uses
        # 0.7*vector + 0.3*1/(1+rank))
        self.vector_weight = float(_env("HYBRID_VECTOR_WEIGHT", "0.7"))
        self.bm25_weight = float(_env("HYBRID_BM25_WEIGHT", "0.3"))

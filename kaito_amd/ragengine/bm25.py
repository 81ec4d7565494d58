"""BM25 (Okapi) lexical retriever — the reference builds one fresh from the
docstore per query (hybrid_retriever.py:104-130); ours keeps incremental
stats per index so queries don't pay a rebuild."""
from __future__ import annotations

import math
import re
from collections import Counter
from typing import Dict, List, Tuple


def tokenize(text: str) -> List[str]:
    return re.findall(r"\w+", text.lower())


class BM25Index:
    def __init__(self, k1: float = 1.5, b: float = 0.75):
        self.k1 = k1
        self.b = b
        self.doc_tokens: Dict[str, Counter] = {}
        self.doc_len: Dict[str, int] = {}
        self.df: Counter = Counter()
        self.total_len = 0

    def add(self, doc_id: str, text: str) -> None:
        if doc_id in self.doc_tokens:
            self.remove(doc_id)
        toks = Counter(tokenize(text))
        self.doc_tokens[doc_id] = toks
        n = sum(toks.values())
        self.doc_len[doc_id] = n
        self.total_len += n
        for t in toks:
            self.df[t] += 1

    def remove(self, doc_id: str) -> None:
        toks = self.doc_tokens.pop(doc_id, None)
        if toks is None:
            return
        self.total_len -= self.doc_len.pop(doc_id, 0)
        for t in toks:
            self.df[t] -= 1
            if self.df[t] <= 0:
                del self.df[t]

    def __len__(self) -> int:
        return len(self.doc_tokens)

    def search(self, query: str, top_k: int = 10) -> List[Tuple[str, float]]:
        if not self.doc_tokens:
            return []
        q = tokenize(query)
        N = len(self.doc_tokens)
        avgdl = self.total_len / max(N, 1)
        scores: Dict[str, float] = {}
        for term in q:
            df = self.df.get(term)
            if not df:
                continue
            idf = math.log(1 + (N - df + 0.5) / (df + 0.5))
            for doc_id, toks in self.doc_tokens.items():
                tf = toks.get(term)
                if not tf:
                    continue
                dl = self.doc_len[doc_id]
                s = idf * tf * (self.k1 + 1) / (
                    tf + self.k1 * (1 - self.b + self.b * dl / avgdl))
                scores[doc_id] = scores.get(doc_id, 0.0) + s
        return sorted(scores.items(), key=lambda kv: -kv[1])[:top_k]

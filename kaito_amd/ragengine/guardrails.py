"""Output guardrails for the RAG service.

Reference parity: presets/ragengine guardrails (LLM-Guard scanners with a
hot-reloading YAML policy, guardrails/reload.py; streaming buffer-window
scan, streaming/buffer_window.py + streaming/guardrails.py). LLM-Guard is
not in this image, so scanners are policy-driven regex/keyword matchers —
the same policy file shape and the same enforcement points.

Policy YAML:
  blocked_patterns: ["(?i)ssn\\s*\\d{3}-\\d{2}-\\d{4}"]
  blocked_keywords: ["secret-project"]
  redactions:
    - pattern: "\\b\\d{16}\\b"
      replacement: "[REDACTED-CARD]"
  action: block | redact     (default block on blocked_*, always redact
                              redactions)
"""
from __future__ import annotations

import os
import re
import threading
from dataclasses import dataclass, field
from typing import List, Optional

import yaml


@dataclass
class Violation:
    kind: str
    match: str


@dataclass
class ScanResult:
    ok: bool
    text: str
    violations: List[Violation] = field(default_factory=list)


class Policy:
    def __init__(self, data: Optional[dict] = None):
        data = data or {}
        self.blocked_patterns = [re.compile(p)
                                 for p in data.get("blocked_patterns", [])]
        self.blocked_keywords = [k.lower()
                                 for k in data.get("blocked_keywords", [])]
        self.redactions = [(re.compile(r["pattern"]),
                            r.get("replacement", "[REDACTED]"))
                           for r in data.get("redactions", [])]
        self.action = data.get("action", "block")
        self.block_message = data.get(
            "block_message", "[output blocked by guardrails policy]")


class PolicyLoader:
    """Hot-reloading policy file (reference guardrails/reload.py): re-reads
    when the file mtime changes."""

    def __init__(self, path: Optional[str], hot_reload: bool = True):
        self.path = path
        self.hot_reload = hot_reload
        self._mtime = 0.0
        self._policy = Policy()
        self._lock = threading.Lock()
        self._load()

    def _load(self):
        if not self.path or not os.path.exists(self.path):
            return
        mtime = os.path.getmtime(self.path)
        if mtime == self._mtime:
            return
        with open(self.path) as f:
            data = yaml.safe_load(f) or {}
        self._policy = Policy(data)
        self._mtime = mtime

    def get(self) -> Policy:
        with self._lock:
            if self.hot_reload:
                self._load()
            return self._policy


class Scanner:
    def __init__(self, loader: PolicyLoader):
        self.loader = loader

    def scan(self, text: str) -> ScanResult:
        pol = self.loader.get()
        violations: List[Violation] = []
        out = text
        for pat, repl in pol.redactions:
            if pat.search(out):
                violations.append(Violation("redaction", pat.pattern))
                out = pat.sub(repl, out)
        low = out.lower()
        for kw in pol.blocked_keywords:
            if kw in low:
                violations.append(Violation("keyword", kw))
        for pat in pol.blocked_patterns:
            m = pat.search(out)
            if m:
                violations.append(Violation("pattern", m.group(0)))
        blocked = any(v.kind in ("keyword", "pattern") for v in violations)
        if blocked and pol.action == "block":
            return ScanResult(False, pol.block_message, violations)
        return ScanResult(not blocked, out, violations)


class BufferWindowScanner:
    """Streaming scan (reference streaming/buffer_window.py): holds back a
    window so matches spanning chunk boundaries are caught; emits cleared
    text; a violation truncates the stream with the block message."""

    def __init__(self, scanner: Scanner, window: int = 64):
        self.scanner = scanner
        self.window = window
        self._buf = ""
        self.blocked = False

    def feed(self, chunk: str) -> str:
        if self.blocked:
            return ""
        self._buf += chunk
        res = self.scanner.scan(self._buf)
        if not res.ok:
            self.blocked = True
            out = res.text  # block message
            self._buf = ""
            return out
        # emit everything except the trailing window (kept for boundary
        # matches); redactions apply to the emitted prefix
        if len(res.text) <= self.window:
            self._buf = res.text
            return ""
        emit = res.text[:-self.window]
        self._buf = res.text[-self.window:]
        return emit

    def flush(self) -> str:
        if self.blocked:
            return ""
        res = self.scanner.scan(self._buf)
        self._buf = ""
        return res.text

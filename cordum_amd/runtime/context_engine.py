"""Context engine: BuildWindow / UpdateMemory over the memory namespace.

Oracle: core/context/engine/service.go:55-288 — modes RAW/CHAT/RAG; chat
history `mem:<id>:events` (list, trimmed to the last 20), RAG chunks
`mem:<id>:chunk:<i>` matched by file_path plus summary `mem:<id>:summary`;
token estimate = len/4; trim-to-budget drops oldest; UpdateMemory appends
user/assistant events. gRPC contract: core/protocol/proto/v1/context.proto.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List

from ..store.memory_store import MemoryStore

MODE_RAW = "raw"
MODE_CHAT = "chat"
MODE_RAG = "rag"

MAX_CHAT_EVENTS = 20


@dataclass
class ModelMessage:
    role: str
    content: str

    def to_dict(self):
        return {"role": self.role, "content": self.content}


@dataclass
class Window:
    messages: List[ModelMessage] = field(default_factory=list)
    input_tokens: int = 0
    output_tokens: int = 0


def estimate_tokens(text: str) -> int:
    return len(text) // 4


class ContextEngine:
    def __init__(self, memory: MemoryStore):
        self.memory = memory

    # -- memory keys -----------------------------------------------------------
    def _events_key(self, memory_id: str) -> str:
        return f"mem:{memory_id}:events"

    def _load_events(self, memory_id: str) -> List[Dict[str, str]]:
        blob = self.memory.get(self._events_key(memory_id))
        if not blob:
            return []
        try:
            events = json.loads(blob)
            return events if isinstance(events, list) else []
        except ValueError:
            return []

    def _save_events(self, memory_id: str, events: List[Dict[str, str]]) -> None:
        self.memory.put(self._events_key(memory_id), json.dumps(events[-MAX_CHAT_EVENTS:]).encode())

    # -- BuildWindow ------------------------------------------------------------
    def build_window(
        self,
        memory_id: str,
        mode: str = MODE_RAW,
        logical_payload: bytes = b"",
        max_input_tokens: int = 8000,
        max_output_tokens: int = 1024,
    ) -> Window:
        mode = (mode or MODE_RAW).lower()
        prompt = ""
        file_path = ""
        try:
            payload = json.loads(logical_payload.decode("utf-8")) if logical_payload else {}
            if isinstance(payload, dict):
                prompt = str(payload.get("prompt", "") or "")
                file_path = str(payload.get("file_path", "") or "")
        except (ValueError, UnicodeDecodeError):
            prompt = logical_payload.decode("utf-8", errors="replace")

        messages: List[ModelMessage] = []
        if mode == MODE_CHAT:
            for ev in self._load_events(memory_id):
                messages.append(ModelMessage(str(ev.get("role", "user")), str(ev.get("content", ""))))
            messages.append(ModelMessage("user", prompt))
        elif mode == MODE_RAG:
            summary = self.memory.get(f"mem:{memory_id}:summary")
            if summary:
                messages.append(ModelMessage("system", summary.decode("utf-8", errors="replace")))
            i = 0
            while True:
                chunk = self.memory.get(f"mem:{memory_id}:chunk:{i}")
                if chunk is None:
                    break
                try:
                    doc = json.loads(chunk)
                except ValueError:
                    doc = {"content": chunk.decode("utf-8", errors="replace")}
                if not file_path or doc.get("file_path", "") == file_path:
                    messages.append(ModelMessage("system", str(doc.get("content", ""))))
                i += 1
            messages.append(ModelMessage("user", prompt))
        else:  # RAW
            messages.append(ModelMessage("user", prompt))

        # trim to budget, dropping oldest first (service.go:279-288); the
        # final user message is always kept
        def total_tokens() -> int:
            return sum(estimate_tokens(m.content) for m in messages)

        while len(messages) > 1 and total_tokens() > max_input_tokens:
            messages.pop(0)
        return Window(messages=messages, input_tokens=total_tokens(), output_tokens=max_output_tokens)

    # -- UpdateMemory --------------------------------------------------------------
    def update_memory(self, memory_id: str, logical_payload: bytes, model_response: bytes,
                      mode: str = MODE_CHAT) -> None:
        prompt = ""
        try:
            payload = json.loads(logical_payload.decode("utf-8")) if logical_payload else {}
            if isinstance(payload, dict):
                prompt = str(payload.get("prompt", "") or "")
        except (ValueError, UnicodeDecodeError):
            prompt = logical_payload.decode("utf-8", errors="replace")
        events = self._load_events(memory_id)
        if prompt:
            events.append({"role": "user", "content": prompt})
        if model_response:
            events.append({"role": "assistant", "content": model_response.decode("utf-8", errors="replace")})
        self._save_events(memory_id, events)

    # -- RAG ingestion helper --------------------------------------------------------
    def put_chunk(self, memory_id: str, index: int, content: str, file_path: str = "") -> None:
        self.memory.put(f"mem:{memory_id}:chunk:{index}",
                        json.dumps({"content": content, "file_path": file_path}).encode())

    def put_summary(self, memory_id: str, summary: str) -> None:
        self.memory.put(f"mem:{memory_id}:summary", summary.encode())

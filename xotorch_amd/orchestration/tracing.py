"""Request tracing: per-request spans, token-group spans, W3C traceparent.

The reference ships an OpenTelemetry tracer that is never imported by the
main path (/root/reference/xotorch/orchestration/tracing.py — latent,
SURVEY.md §5). Here tracing is wired in for real: Node opens a request span
on process_prompt, token-group spans every TOKEN_GROUP_SIZE sampled tokens,
and the traceparent string travels on the wire so multi-host rings correlate.
Spans export to OpenTelemetry when the SDK is importable, else to an
in-memory ring + optional JSONL file (XOT_TRACE_FILE).
"""
from __future__ import annotations

import json
import os
import random
import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Dict, List, Optional

TOKEN_GROUP_SIZE = 10


def _rand_hex(n: int) -> str:
  return "".join(random.choice("0123456789abcdef") for _ in range(n))


@dataclass
class Span:
  name: str
  trace_id: str
  span_id: str
  parent_id: Optional[str]
  start_ns: int
  end_ns: Optional[int] = None
  attributes: Dict[str, object] = field(default_factory=dict)

  @property
  def duration_ms(self) -> Optional[float]:
    if self.end_ns is None:
      return None
    return (self.end_ns - self.start_ns) / 1e6

  def to_dict(self):
    return {
      "name": self.name, "trace_id": self.trace_id, "span_id": self.span_id,
      "parent_id": self.parent_id, "start_ns": self.start_ns, "end_ns": self.end_ns,
      "duration_ms": self.duration_ms, "attributes": self.attributes,
    }


@dataclass
class TraceContext:
  request_id: str
  trace_id: str
  root_span: Optional[Span] = None
  token_count: int = 0
  group_span: Optional[Span] = None


class Tracer:
  def __init__(self, max_spans: int = 4096):
    self._lock = threading.Lock()
    self.finished: deque = deque(maxlen=max_spans)
    self.contexts: Dict[str, TraceContext] = {}
    self.trace_file = os.getenv("XOT_TRACE_FILE")

  # --- W3C traceparent propagation ---

  def inject(self, ctx: TraceContext) -> str:
    parent = ctx.root_span.span_id if ctx.root_span else _rand_hex(16)
    return f"00-{ctx.trace_id}-{parent}-01"

  def extract(self, traceparent: Optional[str], request_id: str) -> TraceContext:
    if traceparent:
      try:
        _, trace_id, parent_id, _ = traceparent.split("-")
        ctx = TraceContext(request_id=request_id, trace_id=trace_id)
        ctx.root_span = self.start_span("remote-segment", ctx, parent_id=parent_id)
        self.contexts[request_id] = ctx
        return ctx
      except ValueError:
        pass
    return self.start_request(request_id)

  # --- span lifecycle ---

  def start_request(self, request_id: str, name: str = "request", **attrs) -> TraceContext:
    ctx = TraceContext(request_id=request_id, trace_id=_rand_hex(32))
    ctx.root_span = self.start_span(name, ctx, attributes=dict(attrs, request_id=request_id))
    with self._lock:
      self.contexts[request_id] = ctx
    return ctx

  def start_span(self, name: str, ctx: TraceContext, parent_id: Optional[str] = None,
                 attributes: Optional[dict] = None) -> Span:
    parent = parent_id or (ctx.root_span.span_id if ctx.root_span else None)
    return Span(name=name, trace_id=ctx.trace_id, span_id=_rand_hex(16),
                parent_id=parent, start_ns=time.perf_counter_ns(),
                attributes=attributes or {})

  def end_span(self, span: Span, **attrs):
    span.end_ns = time.perf_counter_ns()
    span.attributes.update(attrs)
    with self._lock:
      self.finished.append(span)
    if self.trace_file:
      try:
        with open(self.trace_file, "a") as f:
          f.write(json.dumps(span.to_dict()) + "\n")
      except OSError:
        pass

  def handle_token(self, request_id: str, is_finished: bool = False):
    """Group tokens into spans of TOKEN_GROUP_SIZE (reference tracing.py:72-103)."""
    ctx = self.contexts.get(request_id)
    if ctx is None:
      return
    if ctx.group_span is None:
      ctx.group_span = self.start_span(f"tokens[{ctx.token_count}..]", ctx)
    ctx.token_count += 1
    if ctx.token_count % TOKEN_GROUP_SIZE == 0 or is_finished:
      self.end_span(ctx.group_span, tokens=ctx.token_count)
      ctx.group_span = None
    if is_finished:
      self.end_request(request_id)

  def end_request(self, request_id: str, **attrs):
    ctx = self.contexts.pop(request_id, None)
    if ctx is None:
      return
    if ctx.group_span is not None:
      self.end_span(ctx.group_span, tokens=ctx.token_count)
    if ctx.root_span is not None:
      self.end_span(ctx.root_span, total_tokens=ctx.token_count, **attrs)

  def request_spans(self, trace_id: str) -> List[Span]:
    with self._lock:
      return [s for s in self.finished if s.trace_id == trace_id]


tracer = Tracer()

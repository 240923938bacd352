"""Tracing tests: span lifecycle, token grouping, traceparent propagation."""
from xotorch_amd.orchestration.tracing import TOKEN_GROUP_SIZE, Tracer


def test_request_and_token_group_spans():
  t = Tracer()
  ctx = t.start_request("r1")
  for i in range(TOKEN_GROUP_SIZE * 2 + 3):
    t.handle_token("r1", is_finished=False)
  t.handle_token("r1", is_finished=True)
  spans = t.request_spans(ctx.trace_id)
  names = [s.name for s in spans]
  assert sum(1 for n in names if n.startswith("tokens[")) == 3
  assert "request" in names
  root = [s for s in spans if s.name == "request"][0]
  assert root.attributes["total_tokens"] == TOKEN_GROUP_SIZE * 2 + 4
  assert all(s.duration_ms is not None for s in spans)


def test_traceparent_roundtrip():
  t = Tracer()
  ctx = t.start_request("r2")
  header = t.inject(ctx)
  assert header.startswith("00-") and len(header.split("-")) == 4
  t2 = Tracer()
  ctx2 = t2.extract(header, "r2")
  assert ctx2.trace_id == ctx.trace_id


def test_extract_without_header_starts_fresh():
  t = Tracer()
  ctx = t.extract(None, "r3")
  assert ctx.trace_id and ctx.root_span is not None

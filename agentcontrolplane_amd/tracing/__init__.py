from .tracer import Span, Tracer, get_tracer, reconstruct_span_context  # noqa: F401

from .trace import Tracer, dump_global, get_tracer  # noqa: F401

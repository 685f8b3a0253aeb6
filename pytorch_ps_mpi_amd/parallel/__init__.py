from . import async_ps, comm, engines  # noqa: F401

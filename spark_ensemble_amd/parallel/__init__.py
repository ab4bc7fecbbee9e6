from .dist import Comm, get_comm, init_from_env  # noqa: F401

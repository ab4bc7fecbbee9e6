from .dist import Comm, get_comm, init_from_env, set_comm  # noqa: F401

from .env import get_env_int, get_env_float, get_env_bool  # noqa: F401

from .env import env_flag, env_float  # noqa: F401

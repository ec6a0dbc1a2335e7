from .sac import SACEngine  # noqa: F401

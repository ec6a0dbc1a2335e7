from .sac import SACEngine  # noqa: F401
from .care import CAREEngine  # noqa: F401


def create_engine(cfg, device="cpu", precision=None):
    """Variant dispatch: sac/vsac/mtsac -> SACEngine, care -> CAREEngine."""
    if cfg.variant == "care":
        return CAREEngine(cfg, device, precision=precision)
    return SACEngine(cfg, device, precision=precision)

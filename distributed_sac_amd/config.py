"""Config system — reference-compatible cfg-JSON loading.

The reference parses its cfg JSONs with a custom ``json.JSONDecoder`` that
recursively coerces *numeric strings* to int (reference
``LunarLander_Distributed_SAC/src/utils.py:4-20``, ``MT10_Distributed_MTSAC/
src/utils.py:24-27``).  We reproduce those semantics exactly so the shipped
cfg files (``cfg/*.json``) parse to identical dicts, and add a typed overlay
(:class:`SACConfig`) used by the engine.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


class Decoder(json.JSONDecoder):
    """Recursively coerce numeric strings to int while decoding.

    Mirrors reference ``utils.Decoder`` (LunarLander…/src/utils.py:4-20):
    dict values that are strings of digits become ints; nested dicts/lists
    are walked.  JSON numbers (``1e6``, ``3e-4``) pass through unchanged.
    """

    def decode(self, s, **kwargs):
        result = super().decode(s, **kwargs)
        return self._decode(result)

    def _decode(self, o: Any) -> Any:
        if isinstance(o, str):
            try:
                return int(o)
            except ValueError:
                return o
        if isinstance(o, dict):
            return {k: self._decode(v) for k, v in o.items()}
        if isinstance(o, list):
            return [self._decode(v) for v in o]
        return o


def cfg_read(path: str) -> Dict[str, Any]:
    """Load a cfg JSON with reference coercion semantics.

    Mirrors reference ``cfg_read`` (MT10_Distributed_MTSAC/src/utils.py:24-27).
    """
    with open(path, "r") as f:
        return json.loads(f.read(), cls=Decoder)


# ---------------------------------------------------------------------------
# Typed overlay used by the MI355X engine.  Every field defaults to the
# reference's shipped values so a bare cfg dict round-trips losslessly.
# ---------------------------------------------------------------------------

VARIANTS = ("sac", "vsac", "mtsac", "care")


@dataclass
class SACConfig:
    """Normalized view over the five reference cfg schemas.

    The raw dict is kept in :attr:`raw`; unknown keys are preserved so
    round-tripping a reference cfg is lossless.
    """

    variant: str = "sac"                 # sac | vsac | mtsac | care
    device: str = "cuda"
    state_dim: int = 8
    action_dim: int = 2
    action_bound: List[float] = field(default_factory=lambda: [-1.0, 1.0])
    actor_hidden_dim: List[int] = field(default_factory=lambda: [256, 256])
    critic_hidden_dim: List[int] = field(default_factory=lambda: [256, 256])
    num_tasks: int = 1

    buffer_size: int = 1_000_000
    batch_size: int = 256
    reward_scale: float = 1.0
    gamma: float = 0.99
    tau: float = 0.005
    lr_actor: float = 3e-4
    lr_critic: float = 3e-4
    log_alpha: float = 0.0
    update_delay: int = 3
    random_step: int = 5000
    start_memory_len: int = 5000
    max_episode_time: int = 500

    use_weighted_loss: bool = False
    # "corrected" (default): true per-sample task weighting.
    # "reference": reproduce the reference's degenerate (B,)x(B,1)->(B,B)
    # broadcast where the weights cancel (= mean(loss)/B) — strict-parity
    # mode, CPU torch path only.  See docs/PARITY.md "weighted loss".
    weighted_loss_mode: str = "corrected"
    # CARE-only block (reference cfg "encoder"):
    encoder: Optional[Dict[str, Any]] = None
    use_modified_care: bool = False
    state_encoder_tau: float = 0.05

    raw: Dict[str, Any] = field(default_factory=dict)

    @property
    def mtobs_dim(self) -> int:
        """state_dim + one-hot task suffix (MT variants; reference
        MT10_Distributed_MTSAC/src/model.py:20)."""
        if self.variant in ("mtsac", "care"):
            return self.state_dim + self.num_tasks
        return self.state_dim

    @property
    def k(self) -> float:
        """Action scale: (hi - lo) / 2 (reference model.py)."""
        return (self.action_bound[1] - self.action_bound[0]) / 2

    @classmethod
    def from_dict(cls, cfg: Dict[str, Any], variant: str = None) -> "SACConfig":
        """Build from a reference-schema cfg dict (flat LL/VSAC style or the
        nested actor/critic/encoder MT style)."""
        c = cls()
        c.raw = dict(cfg)
        actor = cfg.get("actor", {})
        critic = cfg.get("critic", {})

        def pick(key, *scopes, default=None):
            for s in scopes:
                if key in s:
                    return s[key]
            return default

        c.device = cfg.get("device", c.device)
        c.state_dim = int(pick("state_dim", actor, cfg, default=c.state_dim))
        c.action_dim = int(pick("action_dim", actor, cfg, default=c.action_dim))
        c.action_bound = pick("action_bound", actor, cfg, default=c.action_bound)
        c.actor_hidden_dim = pick("actor_hidden_dim", actor, cfg, default=c.actor_hidden_dim)
        c.critic_hidden_dim = pick("critic_hidden_dim", critic, cfg, default=c.critic_hidden_dim)
        c.lr_actor = float(pick("lr_actor", actor, cfg, default=c.lr_actor))
        c.lr_critic = float(pick("lr_critic", critic, cfg, default=c.lr_critic))

        for key in ("buffer_size", "batch_size", "random_step",
                    "start_memory_len", "update_delay", "max_episode_time",
                    "num_tasks"):
            if key in cfg:
                setattr(c, key, int(cfg[key]))
        for key in ("reward_scale", "gamma", "tau", "log_alpha"):
            if key in cfg:
                setattr(c, key, float(cfg[key]))
        if "use_weighted_loss" in cfg:
            c.use_weighted_loss = bool(cfg["use_weighted_loss"])
        if "weighted_loss_mode" in cfg:
            c.weighted_loss_mode = str(cfg["weighted_loss_mode"])
        if "use_modified_care" in cfg:
            c.use_modified_care = bool(cfg["use_modified_care"])
        if "encoder" in cfg:
            c.encoder = cfg["encoder"]
        if "state_encoder_tau" in cfg:
            c.state_encoder_tau = float(cfg["state_encoder_tau"])
        if c.encoder and "state_encoder_tau" in c.encoder:
            c.state_encoder_tau = float(c.encoder["state_encoder_tau"])

        if variant is not None:
            c.variant = variant
        else:
            # Infer: encoder block => care; num_tasks>1 => mtsac; nested
            # actor block without encoder and num_tasks==1 => vsac; else sac.
            if "encoder" in cfg:
                c.variant = "care"
            elif int(cfg.get("num_tasks", 1)) > 1 and "actor" in cfg:
                c.variant = "mtsac"
            elif "actor" in cfg or "actor_hidden_dim" in cfg:
                c.variant = "vsac" if c.state_dim == 39 else "sac"
            else:
                c.variant = "sac"
        assert c.variant in VARIANTS, c.variant
        return c

    @classmethod
    def from_file(cls, path: str, variant: str = None) -> "SACConfig":
        return cls.from_dict(cfg_read(path), variant)


_CFG_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "cfg")

# Canonical shipped configs (same filenames as the reference cfg/ dir).
CANONICAL_CFGS = {
    "sac": "LunarLanderContinuous-v2_Distributed_SAC_cfg.json",
    "vsac": "MT1_Distributed_VSAC_cfg.json",
    "mtsac": "MT10_Distributed_MTSAC_cfg.json",
    "care": "MT10_Distributed_CARE_cfg.json",
    "mt1_care": "MT1_Distributed_CARE_cfg.json",
}


def load_variant(variant: str, cfg_dir: str = None) -> SACConfig:
    """Load the canonical shipped cfg for a variant."""
    cfg_dir = cfg_dir or _CFG_DIR
    name = CANONICAL_CFGS[variant]
    v = "care" if variant == "mt1_care" else variant
    return SACConfig.from_file(os.path.join(cfg_dir, name), v)

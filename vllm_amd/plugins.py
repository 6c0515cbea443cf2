"""Entry-point plugin loader (role of the reference's vllm/plugins/:
load_general_plugins). Third-party packages register a callable under
the `vllm_amd.plugins` entry-point group; each is invoked once at
engine construction (model registrations, custom logits processors,
platform hooks). Failures are logged and skipped — a broken plugin
must not take the engine down."""

from __future__ import annotations

import logging

logger = logging.getLogger(__name__)

_loaded = False


def _iter_entry_points():
    from importlib.metadata import entry_points

    eps = entry_points()
    if hasattr(eps, "select"):  # py3.10+: SelectableGroups / EntryPoints
        return list(eps.select(group="vllm_amd.plugins"))
    return list(eps.get("vllm_amd.plugins", []))


def load_plugins() -> int:
    """Load every registered plugin once per process; returns how many
    ran (0 on repeat calls)."""
    global _loaded
    if _loaded:
        return 0
    _loaded = True
    n = 0
    for ep in _iter_entry_points():
        try:
            hook = ep.load()
            hook()
            n += 1
            logger.info("loaded plugin %s", ep.name)
        except Exception:  # noqa: BLE001
            logger.exception("plugin %s failed to load; skipping",
                             ep.name)
    return n

"""Compatibility namespace: `moco.builder` / `moco.loader` re-export the
MI355X-native implementations so reference-style imports keep working."""

from . import builder, loader  # noqa: F401

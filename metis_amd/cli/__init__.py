"""CLI entry points, flag-compatible with the reference scripts."""

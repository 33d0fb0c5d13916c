"""Cross-cutting utilities: structured request logging + service metrics."""

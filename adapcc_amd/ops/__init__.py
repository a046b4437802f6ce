"""Native extension build + fused-op wrappers (see csrc/ and fused.py)."""

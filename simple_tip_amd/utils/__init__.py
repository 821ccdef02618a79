"""Shared utilities (tokenizer, logging helpers)."""

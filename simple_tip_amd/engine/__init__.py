"""Experiment engine: handlers, evaluators, ensemble runner."""

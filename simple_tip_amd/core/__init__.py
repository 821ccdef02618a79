"""Core, reusable TIP metric library (framework-independent, unit-tested)."""

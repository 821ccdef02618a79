"""Results aggregation: APFD tables, active-learning tables, statistics."""

"""Language-model training components (dataset + engine)."""

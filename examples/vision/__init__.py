"""Vision training components for the example CLIs."""

"""In-repo test helpers (not shipped in the wheel).

Mirror of the reference's testing/ package strategy:
fork-based multi-process harness over gloo for CPU logic tests,
tiny real models, and a trivial work assignment that exercises every
branch of the preconditioner control flow.
"""

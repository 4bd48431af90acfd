"""Frontends: CLI, gRPC server, C ABI, and the pysonata-compatible Python
API — the four user-facing surfaces of the reference engine
(SURVEY.md §2.1 #9-12)."""

"""Wire-compatible Vizier protos, built at import time (see schema.py)."""

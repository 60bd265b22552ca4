"""Config presets. Each module exports a single ``config`` object
(parity with reference src/configs/*.py)."""

"""Engines (native xGMI / p2p / process-group) and the DDP hook."""

"""Relay/fault coordinator: rank-0 gRPC server + per-rank clients."""

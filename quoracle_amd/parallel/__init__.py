"""Distributed layer: gloo control plane + RCCL-over-xGMI data plane."""

"""Kubelet device-plugin gRPC API (v1beta1) — protocol + server + test kubelet."""

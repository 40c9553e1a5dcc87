"""Cluster state: k8s REST + kubelet clients, pod manager, annotation protocol."""
